"""Multi-process collective tests (gloo, world_size=2, CPU) for the
distributed components: RCCL-path model broadcast and drift all-reduce.
The same code runs backend "nccl" (= RCCL) on the 8-GPU node."""

from __future__ import annotations

import multiprocessing as mp
import os

import numpy as np
import pytest


def _init(rank: int, world: int, port: int):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _broadcast_worker(rank: int, world: int, port: int, npz_path: str, q):
    try:
        import torch.distributed as dist

        from creditcore.pack import PackedModel
        from creditcore.parallel import broadcast_packed

        _init(rank, world, port)
        packed = PackedModel.load(npz_path) if rank == 0 else None
        got = broadcast_packed(packed, device="cpu", src=0)
        # every rank scores identically after broadcast
        from creditcore.ops import cpu_ref
        from creditcore.pack import encode_batch
        from creditcore.data import make_request_batch

        codes, nums = encode_batch(make_request_batch(64, seed=1), got.vocabs)
        out = cpu_ref.score_batch_cpu(got, codes, nums)
        q.put((rank, out["predictions"].tolist(), got.cls_n_trees))
        dist.destroy_process_group()
    except Exception as e:  # surface worker failures to the parent
        q.put((rank, f"ERROR: {type(e).__name__}: {e}", None))


def _drift_worker(rank: int, world: int, port: int, npz_path: str, q):
    try:
        import torch
        import torch.distributed as dist

        from creditcore.pack import PackedModel, encode_batch
        from creditcore.parallel import DriftSync
        from creditcore.ops import cpu_ref
        from creditcore.data import make_request_batch

        _init(rank, world, port)
        packed = PackedModel.load(npz_path)
        ds = DriftSync(packed, device="cpu", n_bins=16)
        # each rank scores a different batch
        recs = make_request_batch(128, seed=10 + rank)
        codes, nums = encode_batch(recs, packed.vocabs)
        nums_imp = cpu_ref.impute_nums(packed, nums)
        cat_hist, _ = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
        ds.accumulate(torch.from_numpy(cat_hist), torch.from_numpy(nums))
        ds.allreduce()
        snap = ds.snapshot()
        q.put((rank, ds.global_.numpy().tolist(), snap["node_feature_drift"]["sex"]))
        dist.destroy_process_group()
    except Exception as e:
        q.put((rank, f"ERROR: {type(e).__name__}: {e}", None))


@pytest.fixture(scope="module")
def packed_npz(packed, tmp_path_factory):
    p = str(tmp_path_factory.mktemp("dist") / "packed.npz")
    packed.save(p)
    return p


def _run_workers(target, packed_npz, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=target, args=(r, 2, port, packed_npz, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, a, b = q.get(timeout=180)
        assert not (isinstance(a, str) and a.startswith("ERROR")), a
        results[rank] = (a, b)
    for p in procs:
        p.join(timeout=60)
    return results


def test_broadcast_packed_two_ranks(packed_npz):
    res = _run_workers(_broadcast_worker, packed_npz, 29511)
    assert res[0][1] == res[1][1]  # same tree count
    np.testing.assert_array_equal(res[0][0], res[1][0])  # identical scores


def test_drift_allreduce_two_ranks(packed_npz):
    res = _run_workers(_drift_worker, packed_npz, 29513)
    # all-reduced histograms identical on both ranks
    np.testing.assert_array_equal(res[0][0], res[1][0])
    # and contain both ranks' rows: numeric bins sum to 2*128 per feature
    from creditcore.pack import PackedModel

    packed = PackedModel.load(packed_npz)
    C = int(packed.ref_cat_offsets[-1])
    g = np.asarray(res[0][0])
    assert g[C : C + 16].sum() == 256


def test_driftsync_single_process(packed):
    """No process group: allreduce degrades to local merge; merge_from sums
    in-process replicas."""
    import torch

    from creditcore.data import make_request_batch
    from creditcore.ops import cpu_ref
    from creditcore.pack import encode_batch
    from creditcore.parallel import DriftSync

    a = DriftSync(packed, device="cpu", n_bins=16)
    b = DriftSync(packed, device="cpu", n_bins=16)
    for ds, seed in ((a, 1), (b, 2)):
        recs = make_request_batch(100, seed=seed)
        codes, nums = encode_batch(recs, packed.vocabs)
        nums_imp = cpu_ref.impute_nums(packed, nums)
        cat_hist, _ = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
        ds.accumulate(torch.from_numpy(cat_hist), torch.from_numpy(nums))
    a.merge_from(b)
    a.allreduce()
    snap = a.snapshot()
    assert snap["rows"] == 200
    assert set(snap["node_feature_drift"].keys()) == {
        f for f in __import__("creditcore.schema", fromlist=["FEATURES"]).FEATURES
    }
    for v in snap["node_feature_drift"].values():
        assert -1e-6 <= v <= 1.0 + 1e-6


def test_driftsync_detects_node_drift(packed):
    import torch

    from creditcore.data import make_request_batch
    from creditcore.ops import cpu_ref
    from creditcore.pack import encode_batch
    from creditcore.parallel import DriftSync

    ds = DriftSync(packed, device="cpu", n_bins=32)
    recs = make_request_batch(512, seed=4, drifted=True)
    codes, nums = encode_batch(recs, packed.vocabs)
    nums_imp = cpu_ref.impute_nums(packed, nums)
    cat_hist, _ = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
    ds.accumulate(torch.from_numpy(cat_hist), torch.from_numpy(nums))
    ds.allreduce()
    drift = ds.snapshot()["node_feature_drift"]
    assert drift["credit_limit"] > 0.99  # shifted numerics
    assert drift["sex"] > 0.99  # all-female batch vs mixed reference


def test_driftsync_cross_process_publish(packed, tmp_path):
    """SO_REUSEPORT workers share drift via published snapshots: two
    instances publishing to the same dir see each other's counts."""
    import os

    import torch

    from creditcore.data import make_request_batch
    from creditcore.ops import cpu_ref
    from creditcore.pack import encode_batch
    from creditcore.parallel import DriftSync

    d = str(tmp_path / "shm")
    a = DriftSync(packed, device="cpu", n_bins=8)
    b = DriftSync(packed, device="cpu", n_bins=8)
    for ds, seed in ((a, 1), (b, 2)):
        recs = make_request_batch(60, seed=seed)
        codes, nums = encode_batch(recs, packed.vocabs)
        nums_imp = cpu_ref.impute_nums(packed, nums)
        hist, _ = cpu_ref.drift_stats_cpu(packed, codes, nums_imp)
        ds.accumulate(torch.from_numpy(hist), torch.from_numpy(nums))
    # emulate two pids: publish b under a different name
    b.publish(d)
    os.replace(
        os.path.join(d, f"drift_{os.getpid()}.npy"),
        os.path.join(d, "drift_99999.npy"),
    )
    a.merge_published(d)
    snap = a.snapshot()
    assert snap["rows"] == 120  # both workers' numeric rows visible


def _lin_broadcast_worker(rank: int, world: int, port: int, npz_path: str, q):
    try:
        import torch.distributed as dist

        from creditcore.pack import PackedModel
        from creditcore.parallel import broadcast_packed

        _init(rank, world, port)
        if rank == 0:
            packed = PackedModel.load(npz_path)
            packed.lin_weight = np.arange(40, dtype=np.float32)
            packed.lin_bias = 0.25
        else:
            packed = None
        got = broadcast_packed(packed, device="cpu", src=0)
        q.put((rank, got.lin_weight.tolist(), float(got.lin_bias)))
        dist.destroy_process_group()
    except Exception as e:
        q.put((rank, f"ERROR: {type(e).__name__}: {e}", None))


def test_broadcast_packed_carries_linear_weights(packed_npz):
    """lin_weight must travel with the broadcast (it was silently dropped
    for non-source ranks before round 2's fix)."""
    res = _run_workers(_lin_broadcast_worker, packed_npz, 29517)
    assert res[0][0] == res[1][0] == list(map(float, range(40)))
    assert res[0][1] == res[1][1] == 0.25


def test_driftsync_load_state_rejects_corruption(packed, tmp_path):
    """Corrupted or layout-mismatched drift-state files start fresh
    (return False) instead of crashing the service at boot."""
    import torch

    from creditcore.parallel import DriftSync

    ds = DriftSync(packed, device="cpu", n_bins=16)
    p = str(tmp_path / "drift.npz")
    # corrupt bytes
    open(p, "wb").write(b"not an npz at all")
    assert ds.load_state(p) is False
    # wrong bin layout
    ds2 = DriftSync(packed, device="cpu", n_bins=8)
    ds2.save_state(p)
    assert ds.load_state(p) is False
    # good round trip
    ds.local += 3
    ds.save_state(p)
    ds3 = DriftSync(packed, device="cpu", n_bins=16)
    assert ds3.load_state(p) is True
    assert torch.equal(ds3.local, ds.local)
