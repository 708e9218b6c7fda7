"""Guard: tools/proof_trace_bench.py's replayed call counts must equal the
CALLCOUNTS.md derivation (sync-step k=20: 45 MSM + 40 iFFT + 40 coset-FFT +
1 extended icoset + 21 constraint-eval passes; aggregation: 13/8/8/1/6).
A drive-by edit to either side shows up here, on CPU, before any GPU run."""
import importlib.util
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _traces():
    spec = importlib.util.spec_from_file_location(
        "proof_trace_bench", os.path.join(REPO, "tools",
                                          "proof_trace_bench.py"))
    mod = importlib.util.module_from_spec(spec)
    # import side effects: the tool inserts paths + imports pywrap; that is
    # CPU-safe (oracle builds without a GPU)
    spec.loader.exec_module(mod)
    return mod.TRACES


def _totals(trace):
    tot = {}
    for _, kind, log_n, count in trace:
        tot[kind] = tot.get(kind, 0) + count
    return tot


def test_step20_counts():
    k, ext_k, trace = _traces()["step20"]
    assert (k, ext_k) == (20, 22)
    t = _totals(trace)
    # CALLCOUNTS.md: A=19, L=3, NZ=11, I=1, j=3
    assert t["msm"] == 19 + 6 + 11 + 3 + 1 + 3 + 2 == 45
    assert t["intt"] == 1 + 19 + 9 + 11 == 40
    assert t["coset"] == 40
    assert t["icoset"] == 1
    assert t["gate"] == 1 + 3 * 3 + 11 == 21


def test_agg_counts():
    for name, n_log, ext in (("agg23", 23, 25), ("committee24", 24, 26)):
        k, ext_k, trace = _traces()[name]
        assert (k, ext_k) == (n_log, ext)
        t = _totals(trace)
        # A=2, L=1, NZ=2, j=3
        assert t["msm"] == 2 + 2 + 2 + 1 + 1 + 3 + 2 == 13
        assert t["intt"] == 8
        assert t["coset"] == 8
        assert t["icoset"] == 1
        assert t["gate"] == 1 + 3 + 2 == 6


def test_keygen_counts_labeled_estimate():
    k, ext_k, trace = _traces()["keygen20"]
    t = _totals(trace)
    assert t["msm"] == 18 + 21 == 39
    assert t["intt"] == 39
    assert t["coset"] == 42
