"""Multi-process (gloo, CPU) tests of the int8 ring all-reduce — the same
code path RCCL uses on GPU (torch.distributed P2P is backend-agnostic)."""
import torch

from tests.conftest import run_distributed


def _ring_worker(rank, world, quant):
    import torch.distributed as dist

    from prime_amd.parallel import ring
    from prime_amd.ops import QBLK

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(42)  # same on all ranks
    n = world * QBLK * 4
    full = torch.randn(world, n)
    mine = full[rank].clone()
    want = full.mean(0)
    if quant:
        ring.ring_allreduce_int8(mine, average=True)
    else:
        ring.allreduce_fp32(mine, average=True)
    err = (mine - want).abs().max().item()
    ref_scale = full.abs().max().item()
    dist.barrier()
    dist.destroy_process_group()
    return err, ref_scale, mine[:8].tolist()


def test_ring_int8_two_ranks():
    outs = run_distributed(_ring_worker, 2, args=(True,))
    for err, scale, _ in outs:
        # per-hop int8 quantization: error ~ W * amax/127
        assert err < scale * 4 / 127 + 1e-6, err
    # all ranks converge to identical values
    assert outs[0][2] == outs[1][2]


def test_ring_int8_four_ranks():
    outs = run_distributed(_ring_worker, 4, args=(True,))
    for err, scale, _ in outs:
        assert err < scale * 8 / 127 + 1e-6, err
    assert outs[0][2] == outs[3][2]


def test_ring_fp32_exact():
    outs = run_distributed(_ring_worker, 2, args=(False,))
    for err, _, _ in outs:
        assert err < 1e-6


def _multi_ring_worker(rank, world):
    import torch.distributed as dist

    from prime_amd.parallel import ring
    from prime_amd.ops import QBLK

    dist.init_process_group("gloo", rank=rank, world_size=world)
    offsets = [o for o in range(1, world) if ring._gcd(o, world) == 1]
    R = len(offsets)
    torch.manual_seed(7)
    n = R * world * QBLK * 2
    full = torch.randn(world, n)
    mine = full[rank].clone()
    want = full.mean(0)
    ring.ring_allreduce_int8_multi(mine, average=True)
    err = (mine - want).abs().max().item()
    scale = full.abs().max().item()
    dist.barrier()
    dist.destroy_process_group()
    return err, scale, mine[:8].tolist()


def test_multi_ring_int8_four_ranks():
    outs = run_distributed(_multi_ring_worker, 4)
    for err, scale, _ in outs:
        assert err < scale * 8 / 127 + 1e-6, err
    assert outs[0][2] == outs[1][2] == outs[3][2]


def test_multi_ring_int8_three_ranks():
    outs = run_distributed(_multi_ring_worker, 3)
    for err, scale, _ in outs:
        assert err < scale * 6 / 127 + 1e-6, err
    assert outs[0][2] == outs[2][2]


def _unaligned_worker(rank, world, multi):
    """Ring with a size NOT divisible by W*QBLK — exercises per-call
    padding (an elastic fleet can land on any live world size)."""
    import torch.distributed as dist

    from prime_amd.parallel import ring
    from prime_amd.ops import QBLK

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(3)
    n = world * QBLK * 2 + 177  # deliberately unaligned
    full = torch.randn(world, n)
    mine = full[rank].clone()
    want = full.mean(0)
    if multi:
        ring.ring_allreduce_int8_multi(mine, average=True)
    else:
        ring.ring_allreduce_int8(mine, average=True)
    err = (mine - want).abs().max().item()
    scale = full.abs().max().item()
    dist.barrier()
    dist.destroy_process_group()
    return err, scale, mine[-8:].tolist()


def test_ring_int8_unaligned_three_ranks():
    outs = run_distributed(_unaligned_worker, 3, args=(False,))
    for err, scale, _ in outs:
        assert err < scale * 6 / 127 + 1e-6, err
    assert outs[0][2] == outs[1][2] == outs[2][2]


def test_multi_ring_int8_unaligned_five_ranks():
    outs = run_distributed(_unaligned_worker, 5, args=(True,))
    for err, scale, _ in outs:
        assert err < scale * 10 / 127 + 1e-6, err
    assert outs[0][2] == outs[4][2]


def test_multi_ring_schedule_all_worlds():
    """Pure-schedule simulation of ring_allreduce_int8_multi's partition
    walk (no torch.distributed): for every W in 2..8 and every coprime
    offset, W simulated ranks exchange partitions by the exact
    _ring_pos schedule; every rank must end holding the full sum in
    every partition (multi-ring is default-on — this is its license)."""
    from prime_amd.parallel.ring import _gcd, _ring_pos

    for W in range(2, 9):
        offsets = [o for o in range(1, W) if _gcd(o, W) == 1]
        for o in offsets:
            # state[r][p] = set of contributing ranks in rank r's copy of
            # partition p; starts as {r}
            state = [[{r} for _ in range(W)] for r in range(W)]
            # reduce-scatter: step s, rank r sends its partition
            # _ring_pos(r,o,W,-s) to (r+o)%W which accumulates into
            # _ring_pos(recv_rank, o, W, -s-1)
            for s in range(W - 1):
                sends = []
                for r in range(W):
                    send_idx = _ring_pos(r, o, W, -s)
                    sends.append((r, (r + o) % W, send_idx, set(state[r][send_idx])))
                for src, dst, idx, contrib in sends:
                    recv_idx = _ring_pos(dst, o, W, -s - 1)
                    assert recv_idx == idx, (W, o, s, "send/recv partition mismatch")
                    state[dst][recv_idx] |= contrib
            for r in range(W):
                own = _ring_pos(r, o, W, 1)
                assert state[r][own] == set(range(W)), (W, o, r, "incomplete reduce")
            # all-gather: step s, rank r forwards the payload received at
            # step s-1 (initially its own reduced partition); receiver
            # writes partition _ring_pos(recv_rank, o, W, -s)
            payload = [state[r][_ring_pos(r, o, W, 1)] for r in range(W)]
            have = [{_ring_pos(r, o, W, 1)} for r in range(W)]
            for s in range(W - 1):
                nxt_payload = [None] * W
                for r in range(W):
                    dst = (r + o) % W
                    recv_idx = _ring_pos(dst, o, W, -s)
                    assert payload[r] == set(range(W)), (W, o, s, r)
                    have[dst].add(recv_idx)
                    nxt_payload[dst] = payload[r]
                payload = nxt_payload
            for r in range(W):
                assert have[r] == set(range(W)), (W, o, r, "incomplete gather")


def _multi_ring_worker_w(rank, world):
    import torch.distributed as dist

    from prime_amd.parallel import ring
    from prime_amd.ops import QBLK

    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(11 + world)
    n = world * QBLK * 3 + 555  # unaligned on purpose
    full = torch.randn(world, n)
    mine = full[rank].clone()
    want = full.mean(0)
    ring.ring_allreduce_int8_multi(mine, average=True)
    err = (mine - want).abs().max().item()
    scale = full.abs().max().item()
    dist.barrier()
    dist.destroy_process_group()
    return err, scale, mine[:4].tolist()


def test_multi_ring_worlds_5_to_8():
    for world in (5, 6, 7, 8):
        outs = run_distributed(_multi_ring_worker_w, world)
        for err, scale, _ in outs:
            assert err < scale * 2 * world / 127 + 1e-6, (world, err)
        assert outs[0][2] == outs[world - 1][2], world
