"""Autotuner: cache persistence + distributed-consistent choice (CPU)."""
import os
import time


def test_autotuner_cache(tmp_path, monkeypatch):
    monkeypatch.setenv("TD_AUTOTUNE_DIR", str(tmp_path))
    from triton_dist_amd.tune import AutoTuner

    calls = {"n": 0}

    def bench(cfg):
        def run():
            calls["n"] += 1
            time.sleep(0.001 * cfg["x"])
        return run

    t = AutoTuner("unit", [{"x": 3}, {"x": 1}, {"x": 2}], warmup=1, iters=2)
    best = t.tune("k1", bench)
    assert best == {"x": 1}
    n_after_first = calls["n"]
    t2 = AutoTuner("unit", [{"x": 3}, {"x": 1}, {"x": 2}], warmup=1, iters=2)
    best2 = t2.tune("k1", bench)
    assert best2 == {"x": 1}
    assert calls["n"] == n_after_first  # served from the JSON cache


def test_autotune_decorator(tmp_path, monkeypatch):
    monkeypatch.setenv("TD_AUTOTUNE_DIR", str(tmp_path))
    from triton_dist_amd.tune import autotune

    @autotune("deco", [{"s": 5}, {"s": 1}], key=lambda n, **kw: str(n))
    def op(n, tune_config=None):
        time.sleep(0.0005 * tune_config["s"])
        return tune_config["s"]

    assert op(7) == 1
    assert op(7) == 1


def test_contextual_autotuner(tmp_path, monkeypatch):
    import time as _time

    monkeypatch.setenv("TD_AUTOTUNE_DIR", str(tmp_path))
    from triton_dist_amd.tune import ContextualAutoTuner

    calls = []

    def make_composite(cfg):
        def run():
            calls.append(cfg["n"])
            _time.sleep(0.0005 * cfg["n"])  # composite cost grows with n
        return run

    t = ContextualAutoTuner("demo", [{"n": 1}, {"n": 4}], warmup=1, iters=3)
    best = t.tune("k0", make_composite)
    assert best == {"n": 1}
    # cached second call: no re-benchmark
    before = len(calls)
    assert t.tune("k0", make_composite) == {"n": 1}
    assert len(calls) == before


def test_perf_model_methods():
    from triton_dist_amd.perf_model import (choose_ag_method,
                                            ll_allgather_time_us,
                                            reduce_scatter_time_us)

    assert choose_ag_method(4 << 10, 8) == "ll"      # tiny: latency wins
    assert choose_ag_method(64 << 20, 8) == "push"   # bulk: wire bytes win
    assert ll_allgather_time_us(1 << 10, 8) < ll_allgather_time_us(1 << 24,
                                                                   8)
    assert reduce_scatter_time_us(1 << 20, 8) > 0


def test_maybe_enable_tunableop_cpu_noop():
    """No GPU here: the shipped TunableOp file must not be activated
    (and a live tuning session must never be overridden)."""
    from triton_dist_amd.tune import maybe_enable_tunableop

    assert maybe_enable_tunableop() is False
    import os
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
    try:
        assert maybe_enable_tunableop() is False
    finally:
        del os.environ["PYTORCH_TUNABLEOP_TUNING"]


def test_maybe_enable_tunableop_loads_shipped_file(monkeypatch):
    """With a (faked) GPU and tunable API present, the loader activates
    the shipped result file read-only: enable(True), tuning off, and
    read_file pointed at autotune_cache/tunableop_gfx950.csv."""
    from pathlib import Path

    import triton_dist_amd.tune as tn

    calls = {}

    class FakeTunable:
        @staticmethod
        def enable(v):
            calls["enable"] = v

        @staticmethod
        def tuning_enable(v):
            calls["tuning"] = v

        @staticmethod
        def read_file(p):
            calls["file"] = p

    monkeypatch.setattr(tn.torch.cuda, "tunable", FakeTunable,
                        raising=False)
    monkeypatch.setattr(tn.torch.cuda, "is_available", lambda: True)
    assert tn.maybe_enable_tunableop() is True
    assert calls["enable"] is True and calls["tuning"] is False
    assert calls["file"].endswith("tunableop_gfx950.csv")
    assert Path(calls["file"]).exists()  # the shipped file is real
