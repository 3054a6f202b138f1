from .cli import main

if __name__ == "__main__":  # don't run on programmatic import
    raise SystemExit(main())
