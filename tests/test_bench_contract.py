"""The driver depends on bench.py's CLI + JSON contract; hold it stable."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract(tmp_path):
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--steps", "30",
         "--warmup", "5", "--p50-probes", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["metric"].startswith("samples/sec")
    assert r["unit"] == "samples/s"
    assert r["steps"] == 30 and r["warmup"] == 5
    assert r["higher_is_better"] is True and r["scaling"] == "weak"
    assert r["data"] == "synthetic"
    cfg = r["config"]
    for key in ("model", "global_batch", "parallelism", "comm", "engine"):
        assert key in cfg, key
    assert cfg["model"] == "Linear(20,1)"
    assert r["value"] > 0 and r["ms_per_step"] > 0


def test_bench_defaults_finish_quickly():
    # "with no flags it must default to N=1 and a K/W that finish within
    # minutes" — run the real defaults end to end
    out = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py")],
                         cwd=ROOT, capture_output=True, text=True,
                         timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    r = json.loads(out.stdout.strip().splitlines()[-1])
    assert r["steps"] == 2000


def test_shards_for_block_matches_per_epoch_concat():
    # CPU fallback of the multi-epoch gather: block e of shards_for_block
    # must equal _gather(e0+e) exactly (the GPU kernel path is held to the
    # same contract bitwise in tests/test_epoch_block_gpu.py)
    import torch

    sys.path.insert(0, ROOT)
    import bench

    d = bench.DeviceData(256, rank=1, world=2, batch=8,
                         device=torch.device("cpu"))
    xs_b, ts_b = d.shards_for_block(3, 4)
    per = d.per_rank
    assert xs_b.shape == (4 * per, 20) and ts_b.shape == (4 * per, 1)
    for e in range(4):
        xs, ts = d._gather(3 + e)
        assert torch.equal(xs_b[e * per:(e + 1) * per], xs)
        assert torch.equal(ts_b[e * per:(e + 1) * per], ts)


class _FakeData:
    """Records which shard buffers are produced; buffers are markers."""

    def __init__(self, spe):
        self.bound_epoch = -1
        self.bound_block = -1
        self.spe = spe
        self.gathers = []

    def shard_for(self, e):
        self.gathers.append(("epoch", e))
        return ("xs", e, 1), ("ts", e, 1)

    def shards_for_block(self, e0, epochs):
        self.gathers.append(("block", e0, epochs))
        return ("xs", e0, epochs), ("ts", e0, epochs)


class _FakeEngine:
    """Emulates the persistent engine's shard-bound deferral contract:
    bind flushes; sequential step indices defer; a non-sequential index
    flushes and restarts; flush launches one kernel covering the pending
    contiguous index range of the CURRENTLY BOUND buffer."""

    def __init__(self, max_defer=1024):
        self.max_defer = max_defer
        self.bound = None
        self.lo = self.hi = 0
        self.launches = []  # (buffer, index_lo, index_hi)

    def bind(self, xs, ts, batch):
        self.flush()
        self.bound = xs
        self.lo = self.hi = 0

    def step_shard(self, i):
        if i == self.hi:
            self.hi += 1
            if self.hi - self.lo >= self.max_defer:
                self.flush()
            return
        self.flush()
        self.lo, self.hi = i, i + 1

    def flush(self):
        if self.hi > self.lo:
            self.launches.append((self.bound, self.lo, self.hi))
            self.lo = self.hi


def _replay(spe, eb, segments, max_defer=1024):
    """Run drive_shard_bound over (start, n) segments; return the flat
    list of (epoch, in-epoch step) pairs actually executed, in order."""
    import torch  # noqa: F401  (bench imports torch at module load)

    sys.path.insert(0, ROOT)
    import bench

    data = _FakeData(spe)
    eng = _FakeEngine(max_defer)
    for start, n in segments:
        bench.drive_shard_bound(data, spe, 32, eb, eng.bind,
                                eng.step_shard, start, n)
    eng.flush()
    executed = []
    for buf, lo, hi in eng.launches:
        assert buf is not None, "stepped before any bind"
        kind, e0, epochs = buf
        for idx in range(lo, hi):
            e_loc, i = divmod(idx, spe)
            assert e_loc < epochs, "step index beyond the bound buffer"
            executed.append((e0 + e_loc, i))
    return executed, data, eng


def _expected(spe, segments):
    out = []
    for start, n in segments:
        out += [divmod(s, spe) for s in range(start, start + n)]
    return out


def test_drive_shard_bound_block_path_covers_every_step_once():
    # the driver's 8-GPU shape: spe=8, default eb = 1024//8 = 128,
    # warmup 500 then a timed region starting mid-block
    spe, eb = 8, 128
    segments = [(0, 500), (500, 3000)]
    executed, data, eng = _replay(spe, eb, segments)
    assert executed == _expected(spe, segments)
    # every gather is a whole block, sequential, gathered exactly once
    assert data.gathers == [("block", b * eb, eb)
                            for b in range(len(data.gathers))]
    # deferral really spans epochs: full blocks launch as ONE kernel
    full = [l for l in eng.launches if l[2] - l[1] == eb * spe]
    assert len(full) >= 2


def test_drive_shard_bound_world1_shape_and_max_defer():
    spe, eb = 64, 16  # world-1 default; block == max_defer exactly
    segments = [(0, 2000), (2000, 5000)]
    executed, data, eng = _replay(spe, eb, segments)
    assert executed == _expected(spe, segments)
    assert max(hi - lo for _, lo, hi in eng.launches) <= 1024


def test_drive_shard_bound_eb1_matches_per_epoch():
    spe = 8
    segments = [(0, 100), (100, 60)]
    executed, data, eng = _replay(spe, 1, segments)
    assert executed == _expected(spe, segments)
    assert all(g[0] == "epoch" for g in data.gathers)
    epochs_gathered = [g[1] for g in data.gathers]
    assert epochs_gathered == sorted(set(epochs_gathered))


def test_drive_shard_bound_odd_block_vs_defer_window():
    # eb*spe NOT a multiple of max_defer: engine auto-flushes mid-block
    executed, data, eng = _replay(8, 100, [(0, 3000)], max_defer=512)
    assert executed == _expected(8, [(0, 3000)])


def test_drive_shard_bound_property_fuzz():
    # randomized spe/eb/max_defer/segment boundaries: the executed
    # (epoch, step) stream must always equal the requested one, and no
    # launch may index past its bound buffer (asserted inside _replay)
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=60, deadline=None)
    @given(st.integers(1, 96), st.integers(1, 160), st.integers(1, 2048),
           st.lists(st.integers(1, 700), min_size=1, max_size=4))
    def run(spe, eb, md, seg_lens):
        segments, start = [], 0
        for n in seg_lens:
            segments.append((start, n))
            start += n
        executed, _, _ = _replay(spe, eb, segments, max_defer=md)
        assert executed == _expected(spe, segments)

    run()


def test_bench_world2_gloo_driver_form():
    # the driver's multi-GPU launch form, on a 2-rank CPU/gloo world:
    # rank 0 must print ONE whole-job JSON line (value aggregated over
    # ranks, elapsed = max over ranks)
    port = 29500 + os.getpid() % 500
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "20", "--warmup", "4",
         "--p50-probes", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-1000:]
    r = json.loads(lines[0])
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["comm"] == "gloo-cpu"
    assert r["value"] > 0


def test_graphed_chunk_decomposition_properties():
    """GraphedAutogradStep.chunks: every run length decomposes exactly,
    largest-first, with the minimal greedy tail (pure logic, CPU)."""
    from hypothesis import given, strategies as st

    from mi355x_ddp.engine import GraphedAutogradStep

    eng = GraphedAutogradStep.__new__(GraphedAutogradStep)
    eng.chunk_sizes = (64, 8, 1)

    @given(st.integers(min_value=1, max_value=5000))
    def check(n):
        cs = eng.chunks(n)
        assert sum(cs) == n                      # covers exactly
        assert all(c in (64, 8, 1) for c in cs)  # only known graphs
        assert cs == sorted(cs, reverse=True)    # largest-first
        # greedy is optimal for divisible chunk ladders (64 = 8*8):
        assert cs.count(8) < 8 and cs.count(1) < 8

    check()

    eng.chunk_sizes = (5, 1)  # non-divisible ladder still covers exactly
    assert eng.chunks(13) == [5, 5, 1, 1, 1]
