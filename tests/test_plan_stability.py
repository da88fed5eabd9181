"""Plan-stability check (dev/auron-it PlanStabilityChecker parity).

Each TPC-DS query's physical plan is serialized (plan/serde msgpack) and
fingerprinted by its operator/expression shape. The golden file pins the
shapes; an unintended change to any query plan fails here. To refresh
after an INTENTIONAL plan change:
    python tests/test_plan_stability.py --refresh
"""
import hashlib
import json
import os

import pytest

GOLDEN = os.path.join(os.path.dirname(__file__), "plan_stability.json")


def _shape(node):
    """Structural fingerprint: operator tree with expr class names (no
    literals, so datagen/catalog tweaks don't churn it)."""
    from auron_amd.plan.nodes import PlanNode

    def expr_sig(e):
        return type(e).__name__

    sig = {"op": type(node).__name__, "children": [_shape(c) for c in node.children()]}
    for f, v in vars(node).items():
        if isinstance(v, PlanNode) or f in ("batches",):
            continue
        if isinstance(v, list) and v and hasattr(v[0], "__class__") and \
                v[0].__class__.__module__.startswith("auron_amd"):
            sig[f] = [expr_sig(x) for x in v]
    return sig


def _fingerprints():
    from auron_amd import AuronSession
    from auron_amd.tpcds import datagen
    from auron_amd.tpcds.queries import QUERIES, Catalog

    root = os.path.join(os.path.dirname(__file__), "..", ".tpcds_cache")
    datagen.write_dataset(root, 0.01)
    s = AuronSession()
    cat = Catalog(root, 0.01)
    out = {}
    for qn in sorted(QUERIES):
        plan = QUERIES[qn](cat, s)
        blob = json.dumps(_shape(plan), sort_keys=True).encode()
        out[qn] = hashlib.sha256(blob).hexdigest()[:16]
    return out


def test_plan_stability():
    if not os.path.exists(GOLDEN):
        pytest.skip("golden file missing; run --refresh")
    golden = json.load(open(GOLDEN))
    got = _fingerprints()
    changed = {q for q in golden if golden.get(q) != got.get(q)}
    assert not changed, (
        f"physical plans changed for {sorted(changed)}; if intentional, "
        "refresh with: python tests/test_plan_stability.py --refresh")
    assert set(got) == set(golden), "query set changed; refresh golden file"


if __name__ == "__main__":
    import sys

    if "--refresh" in sys.argv:
        fp = _fingerprints()
        json.dump(fp, open(GOLDEN, "w"), indent=1, sort_keys=True)
        print(f"wrote {GOLDEN} ({len(fp)} plans)")
