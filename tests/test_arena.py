import torch
import torch.nn as nn

from acco_amd.engine import arena


def tiny_model():
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 4))


def test_param_aliasing_and_values():
    m = tiny_model()
    before = [p.detach().clone() for p in m.parameters()]
    flat = arena.flatten_params(m, torch.float32, torch.device("cpu"), pad_to=256)
    assert flat.numel() % 256 == 0
    for p, b in zip(m.parameters(), before):
        assert torch.equal(p.detach(), b)
    grads = arena.attach_grad_arena(m, torch.float32, torch.device("cpu"), pad_to=256)
    assert arena.check_aliasing(m, flat, grads)


def test_grad_accumulates_into_arena():
    m = tiny_model()
    flat = arena.flatten_params(m, torch.float32, torch.device("cpu"))
    grads = arena.attach_grad_arena(m, torch.float32, torch.device("cpu"))
    x = torch.randn(4, 8)
    m(x).sum().backward()
    g1 = grads.clone()
    assert g1.abs().sum() > 0
    m(x).sum().backward()
    assert torch.allclose(grads, 2 * g1, atol=1e-6)
    # arena writes propagate to model params (views)
    flat.zero_()
    for p in m.parameters():
        assert torch.all(p == 0)


def test_flat_update_changes_forward():
    m = tiny_model()
    flat = arena.flatten_params(m, torch.float32, torch.device("cpu"))
    x = torch.randn(2, 8)
    y0 = m(x)
    flat.mul_(0.5)
    y1 = m(x)
    assert not torch.allclose(y0, y1)


def test_shard_spec_world8_llama1b_size():
    """Driver runs N=1,2,4,8; verify the bucket-major layout invariants at
    world=8 with the flagship parameter count."""
    from acco_amd.parallel.comm import ShardSpec

    n = 1_076_168_704          # llama-1b live numel (bench.py)
    for world in (1, 2, 4, 8):
        spec = ShardSpec.build(n, world, buckets=8)
        assert spec.seg % 256 == 0
        assert spec.total >= n
        assert spec.total == spec.nb * world * spec.seg
        # every (bucket, rank) segment contiguous and inside the buffer
        last = 0
        for b in range(spec.nb):
            for r in range(world):
                off = (b * world + r) * spec.seg
                assert off == last
                last = off + spec.seg
        assert last == spec.total
