"""Find and print the first fact-probe kernel/reference mismatch."""
import sys

import numpy as np
import torch

sys.path.insert(0, ".")
from vainplex_openclaw_amd.ops import gpu as g
from vainplex_openclaw_amd.pipeline.synth import default_facts, synthetic_batch

facts = default_facts()
batch = synthetic_batch(1024, seed=13)
b, o = g.pack_messages(batch.messages)
claims = g.dfa_scan(b, o, "claims")
tk, tv, ph, pw = g.build_fact_table(facts)
dev = b.device
v, c = g.fact_probe(b, o, claims, tk.to(dev), tv.to(dev), ph.to(dev), pw)
cm = [int(m) for m in claims.cpu().numpy().view(np.uint64)]
rv, rc = g.reference_fact_probe(batch.messages, cm, facts)
gv = v.cpu().numpy()
gc = c.cpu().numpy()
bad = np.nonzero((gv != rv) | (gc != rc))[0]
print("mismatches:", len(bad), "of", len(batch.messages))
for i in bad[:5]:
    print("-----")
    print("msg:", batch.messages[i])
    print("claims mask:", hex(cm[i]))
    print("gpu v/c:", gv[i], gc[i], " ref v/c:", rv[i], rc[i])
    toks = g._tokenize_fact(batch.messages[i])
    print("tokens:", toks[:40])
