"""Language packs (10 languages) for cortex signal extraction.

Parity target: cortex `src/patterns/lang-*.ts` — per language: decision /
close / wait / topic regexes, topic blacklist, high-impact keywords, 5
mood regexes (frustrated / excited / tense / productive / exploratory),
noise prefixes. Languages: en de es fr it ja ko pt ru zh
(`src/patterns/registry.ts`).

These packs are authored for this framework (same capability surface, own
wording); the GPU batch path compiles the same families into DFAs.
"""

from __future__ import annotations

from typing import Dict, List

# Each pack: {code, name, patterns:{decision,close,wait,topic}, topic_blacklist,
#             high_impact, moods:{...}, noise_prefixes}

PACKS: Dict[str, dict] = {
    "en": {
        "name": "English",
        "patterns": {
            "decision": [r"\b(?:decided|decision|agreed|let'?s do|the plan is|approach:|we(?:'ll| will) go with)"],
            "close": [
                r"(?:^|\s)(?:is |it's |that's |all )?(?:done|fixed|solved|closed|resolved)(?:\s|[.!]|$)",
                r"(?:^|\s)(?:it |that )works(?:\s|[.!]|$)",
                r"✅",
            ],
            "wait": [r"(?:waiting for|blocked by|blocked on|need(?:s)?\s+\w+\s+first|on hold until)"],
            "topic": [r"(?:back to|now about|regarding|let's (?:talk about|discuss|look at)|switching to)\s+(?:the\s+)?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": [
            "it", "that", "this", "the", "them", "what", "which", "there",
            "nothing", "something", "everything", "me", "you", "him", "her",
            "us", "today", "tomorrow", "yesterday",
        ],
        "high_impact": [
            "architecture", "security", "migration", "delete", "production",
            "deploy", "breaking", "major", "critical", "strategy", "budget", "contract",
        ],
        "moods": {
            "frustrated": r"(?:fuck|shit|damn|sucks|annoying|broken again)",
            "excited": r"(?:nice|awesome|brilliant|sick|great news|love it)",
            "tense": r"(?:careful|risky|urgent|deadline|asap)",
            "productive": r"(?:done|fixed|works|deployed|shipped|merged)",
            "exploratory": r"(?:what if|idea|maybe|experiment|let's try)",
        },
        "noise_prefixes": ["i", "we", "he", "she", "it", "nothing", "something"],
    },
    "de": {
        "name": "Deutsch",
        "patterns": {
            "decision": [r"\b(?:entschieden|entscheidung|beschlossen|einigen wir uns|der plan ist|machen wir so|vorgehen:)"],
            "close": [
                r"(?:^|\s)(?:ist |das ist |alles )?(?:erledigt|fertig|gelöst|behoben|abgeschlossen)(?:\s|[.!]|$)",
                r"(?:^|\s)(?:es |das )funktioniert(?:\s|[.!]|$)",
                r"✅",
            ],
            "wait": [r"(?:warte[tn]? auf|blockiert durch|braucht zuerst|erst wenn)"],
            "topic": [r"(?:zurück zu|jetzt zu|bezüglich|lass uns über|thema)\s+(?:de[mnr]\s+|die\s+|das\s+)?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["es", "das", "dies", "der", "die", "was", "dort", "nichts", "etwas", "alles", "ich", "du", "heute", "morgen", "gestern"],
        "high_impact": ["architektur", "sicherheit", "migration", "löschen", "produktion", "deployment", "kritisch", "strategie", "budget", "vertrag"],
        "moods": {
            "frustrated": r"(?:mist|verdammt|scheiße|nervt|kaputt)",
            "excited": r"(?:super|klasse|genial|großartig|perfekt)",
            "tense": r"(?:vorsicht|riskant|dringend|frist)",
            "productive": r"(?:erledigt|behoben|funktioniert|deployed|fertig)",
            "exploratory": r"(?:was wäre wenn|idee|vielleicht|experiment|probieren)",
        },
        "noise_prefixes": ["ich", "wir", "er", "sie", "es", "nichts", "etwas"],
    },
    "es": {
        "name": "Español",
        "patterns": {
            "decision": [r"\b(?:decidido|decidimos|decisión|acordado|acordamos|el plan es|vamos a hacer|enfoque:)"],
            "close": [r"(?:^|\s)(?:está |todo )?(?:hecho|listo|resuelto|arreglado|cerrado|terminado)(?:\s|[.!]|$)", r"(?:^|\s)funciona(?:\s|[.!]|$)", r"✅"],
            "wait": [r"(?:esperando a|esperando por|bloqueado por|necesita primero|en espera)"],
            "topic": [r"(?:volviendo a|ahora sobre|respecto a|hablemos de|miremos)\s+(?:el\s+|la\s+|los\s+)?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["eso", "esto", "el", "la", "que", "nada", "algo", "todo", "hoy", "mañana", "ayer"],
        "high_impact": ["arquitectura", "seguridad", "migración", "borrar", "producción", "despliegue", "crítico", "estrategia", "presupuesto", "contrato"],
        "moods": {
            "frustrated": r"(?:maldición|mierda|joder|molesto|roto)",
            "excited": r"(?:genial|increíble|brillante|excelente|perfecto)",
            "tense": r"(?:cuidado|arriesgado|urgente|plazo)",
            "productive": r"(?:hecho|arreglado|funciona|desplegado|terminado)",
            "exploratory": r"(?:y si|idea|quizás|experimento|probemos)",
        },
        "noise_prefixes": ["yo", "nosotros", "él", "ella", "eso", "nada", "algo"],
    },
    "fr": {
        "name": "Français",
        "patterns": {
            "decision": [r"\b(?:décidé|décision|convenu|d'accord pour|le plan est|on va faire|approche\s*:)"],
            "close": [r"(?:^|\s)(?:c'est |tout est )?(?:fait|réglé|résolu|corrigé|terminé|fermé)(?:\s|[.!]|$)", r"(?:^|\s)ça (?:marche|fonctionne)(?:\s|[.!]|$)", r"✅"],
            "wait": [r"(?:en attente de|attend|bloqué par|besoin d'abord|en pause jusqu)"],
            "topic": [r"(?:revenons à|maintenant sur|concernant|parlons de|regardons)\s+(?:le\s+|la\s+|les\s+|l')?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["ça", "cela", "le", "la", "que", "rien", "quelque", "tout", "aujourd'hui", "demain", "hier"],
        "high_impact": ["architecture", "sécurité", "migration", "supprimer", "production", "déploiement", "critique", "stratégie", "budget", "contrat"],
        "moods": {
            "frustrated": r"(?:merde|zut|putain|agaçant|cassé)",
            "excited": r"(?:génial|super|brillant|excellent|parfait)",
            "tense": r"(?:attention|risqué|urgent|délai)",
            "productive": r"(?:fait|réglé|fonctionne|déployé|terminé)",
            "exploratory": r"(?:et si|idée|peut-être|expérience|essayons)",
        },
        "noise_prefixes": ["je", "nous", "il", "elle", "ça", "rien"],
    },
    "it": {
        "name": "Italiano",
        "patterns": {
            "decision": [r"\b(?:deciso|decisione|concordato|il piano è|facciamo così|approccio:)"],
            "close": [r"(?:^|\s)(?:è |tutto )?(?:fatto|risolto|sistemato|chiuso|completato)(?:\s|[.!]|$)", r"(?:^|\s)funziona(?:\s|[.!]|$)", r"✅"],
            "wait": [r"(?:in attesa di|aspettando|bloccato da|serve prima|in sospeso)"],
            "topic": [r"(?:torniamo a|ora su|riguardo a|parliamo di|guardiamo)\s+(?:il\s+|la\s+|lo\s+|l')?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["ciò", "questo", "il", "la", "che", "niente", "qualcosa", "tutto", "oggi", "domani", "ieri"],
        "high_impact": ["architettura", "sicurezza", "migrazione", "eliminare", "produzione", "deploy", "critico", "strategia", "budget", "contratto"],
        "moods": {
            "frustrated": r"(?:accidenti|merda|cavolo|fastidioso|rotto)",
            "excited": r"(?:fantastico|grandioso|brillante|ottimo|perfetto)",
            "tense": r"(?:attenzione|rischioso|urgente|scadenza)",
            "productive": r"(?:fatto|risolto|funziona|deployato|completato)",
            "exploratory": r"(?:e se|idea|forse|esperimento|proviamo)",
        },
        "noise_prefixes": ["io", "noi", "lui", "lei", "ciò", "niente"],
    },
    "pt": {
        "name": "Português",
        "patterns": {
            "decision": [r"\b(?:decidido|decidimos|decisão|combinado|acordamos|o plano é|vamos fazer|abordagem:)"],
            "close": [r"(?:^|\s)(?:está |tudo )?(?:feito|pronto|resolvido|corrigido|fechado|concluído)(?:\s|[.!]|$)", r"(?:^|\s)funciona(?:\s|[.!]|$)", r"✅"],
            "wait": [r"(?:esperando por|aguardando|bloqueado por|precisa primeiro|em espera)"],
            "topic": [r"(?:voltando a|agora sobre|sobre o|falemos de|vejamos)\s+(?:o\s+|a\s+|os\s+)?(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["isso", "isto", "o", "a", "que", "nada", "algo", "tudo", "hoje", "amanhã", "ontem"],
        "high_impact": ["arquitetura", "segurança", "migração", "apagar", "produção", "deploy", "crítico", "estratégia", "orçamento", "contrato"],
        "moods": {
            "frustrated": r"(?:droga|merda|caramba|irritante|quebrado)",
            "excited": r"(?:ótimo|incrível|brilhante|excelente|perfeito)",
            "tense": r"(?:cuidado|arriscado|urgente|prazo)",
            "productive": r"(?:feito|resolvido|funciona|implantado|concluído)",
            "exploratory": r"(?:e se|ideia|talvez|experimento|vamos tentar)",
        },
        "noise_prefixes": ["eu", "nós", "ele", "ela", "isso", "nada"],
    },
    "ru": {
        "name": "Русский",
        "patterns": {
            "decision": [r"\b(?:решено|решили|решение|договорились|план такой|будем делать|подход:)"],
            "close": [r"(?:^|\s)(?:всё |все )?(?:готово|сделано|решено|исправлено|закрыто|завершено)(?:\s|[.!]|$)", r"(?:^|\s)работает(?:\s|[.!]|$)", r"✅"],
            "wait": [r"(?:ждём|ожидаем|заблокировано|сначала нужно|в ожидании)"],
            "topic": [r"(?:вернёмся к|теперь о|насчёт|давайте обсудим|посмотрим на)\s+(\w[\w\s-]{3,40})"],
        },
        "topic_blacklist": ["это", "то", "что", "ничего", "что-то", "всё", "сегодня", "завтра", "вчера"],
        "high_impact": ["архитектура", "безопасность", "миграция", "удалить", "продакшн", "деплой", "критично", "стратегия", "бюджет", "контракт"],
        "moods": {
            "frustrated": r"(?:чёрт|блин|бесит|сломалось|достало)",
            "excited": r"(?:отлично|супер|класс|великолепно|круто)",
            "tense": r"(?:осторожно|рискованно|срочно|дедлайн)",
            "productive": r"(?:готово|сделано|работает|задеплоено|завершено)",
            "exploratory": r"(?:а что если|идея|может быть|эксперимент|попробуем)",
        },
        "noise_prefixes": ["я", "мы", "он", "она", "это", "ничего"],
    },
    "ja": {
        "name": "日本語",
        "patterns": {
            "decision": [r"(?:決定|決めました|合意|計画は|方針:|にします)"],
            "close": [r"(?:完了|終わりました|解決|修正済み|できました|直りました)", r"✅"],
            "wait": [r"(?:待っています|待ち|ブロックされ|先に必要|保留中)"],
            "topic": [r"(?:の件|について|に戻りましょう|を話しましょう)\s*(\w[\w\s-]{2,40})?"],
        },
        "topic_blacklist": ["それ", "これ", "あれ", "何", "今日", "明日", "昨日"],
        "high_impact": ["アーキテクチャ", "セキュリティ", "移行", "削除", "本番", "デプロイ", "重大", "戦略", "予算", "契約"],
        "moods": {
            "frustrated": r"(?:くそ|最悪|イライラ|壊れた)",
            "excited": r"(?:すごい|素晴らしい|最高|いいね)",
            "tense": r"(?:注意|危険|緊急|締め切り)",
            "productive": r"(?:完了|修正|動きます|デプロイ済み)",
            "exploratory": r"(?:もし|アイデア|かもしれない|実験|試してみ)",
        },
        "noise_prefixes": ["私", "僕", "彼", "彼女", "それ"],
    },
    "ko": {
        "name": "한국어",
        "patterns": {
            "decision": [r"(?:결정|결정했|합의|계획은|방침:|하기로 했)"],
            "close": [r"(?:완료|끝났|해결|수정됨|됐습니다|고쳤)", r"✅"],
            "wait": [r"(?:기다리는 중|대기 중|차단됨|먼저 필요|보류)"],
            "topic": [r"(?:관련해서|에 대해|다시 돌아가서|이야기해 봅시다)\s*(\w[\w\s-]{2,40})?"],
        },
        "topic_blacklist": ["그것", "이것", "저것", "무엇", "오늘", "내일", "어제"],
        "high_impact": ["아키텍처", "보안", "마이그레이션", "삭제", "프로덕션", "배포", "치명적", "전략", "예산", "계약"],
        "moods": {
            "frustrated": r"(?:젠장|짜증|고장|최악)",
            "excited": r"(?:대박|멋지|훌륭|최고)",
            "tense": r"(?:조심|위험|긴급|마감)",
            "productive": r"(?:완료|수정|작동|배포됨)",
            "exploratory": r"(?:만약|아이디어|아마|실험|해보자)",
        },
        "noise_prefixes": ["나", "우리", "그", "그녀", "그것"],
    },
    "zh": {
        "name": "中文",
        "patterns": {
            "decision": [r"(?:决定|已决定|同意|计划是|方案:|就这么办)"],
            "close": [r"(?:完成|搞定|解决了|修好了|已关闭|好了)", r"✅"],
            "wait": [r"(?:等待|等着|被阻塞|需要先|暂停中)"],
            "topic": [r"(?:回到|关于|现在说说|我们讨论一下)\s*(\w[\w\s-]{2,40})?"],
        },
        "topic_blacklist": ["这个", "那个", "什么", "今天", "明天", "昨天"],
        "high_impact": ["架构", "安全", "迁移", "删除", "生产", "部署", "严重", "战略", "预算", "合同"],
        "moods": {
            "frustrated": r"(?:靠|糟糕|烦人|坏了|气死)",
            "excited": r"(?:太棒|厉害|优秀|真好)",
            "tense": r"(?:小心|有风险|紧急|截止)",
            "productive": r"(?:完成|修好|可以用|已部署)",
            "exploratory": r"(?:如果|想法|也许|实验|试试)",
        },
        "noise_prefixes": ["我", "我们", "他", "她", "这"],
    },
}


def language_codes() -> List[str]:
    return list(PACKS.keys())
