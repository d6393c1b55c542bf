"""CPU unit tests for the MFMA weight-packing relayouts.

The kernels read B-fragments as vectorized loads; the mirrors pre-permute
weights so a lane's K-slice is contiguous (docs/DESIGN.md §2). These
tests pin the exact layout so a packing change cannot silently desync
from the kernel's indexing (which is only exercised by gpu tests).
"""
import torch

from rlgpuschedule_amd.models.hip_actor import HipActorMLP


def test_pack_f32_layout_and_bijectivity():
    K, N = 64, 5
    wt = torch.arange(K * N, dtype=torch.float32).view(K, N)
    packed = HipActorMLP._pack(wt)          # [K/16, 4(s), N, 4(kk)]
    assert packed.shape == (K // 16, 4, N, 4)
    flat = packed.contiguous().flatten()
    for k in range(K):
        c, rem = divmod(k, 16)
        kk, s = divmod(rem, 4)
        for n in range(N):
            # kernel-side address: chunk c, then (s, n, kk) row-major
            j = ((c * 4 + s) * N + n) * 4 + kk
            assert flat[j] == wt[k, n], (k, n)
    # bijective: inverse permute recovers the original
    inv = packed.permute(0, 3, 1, 2).reshape(K, N)
    assert torch.equal(inv, wt)


def test_pack8_bf16_layout_and_bijectivity():
    from rlgpuschedule_amd.models.hip_actor import HipActorMLPBf16
    K, N = 64, 3
    wt = torch.arange(K * N, dtype=torch.float32).view(K, N)
    packed = HipActorMLPBf16._pack8(wt)     # [K/32, 4(s8?), N, 8]
    assert packed.shape == (K // 32, 4, N, 8)
    flat = packed.contiguous().flatten()
    for k in range(K):
        c, rem = divmod(k, 32)
        s, j8 = divmod(rem, 8)
        for n in range(N):
            j = ((c * 4 + s) * N + n) * 8 + j8
            assert flat[j] == wt[k, n], (k, n)
    inv = packed.permute(0, 1, 3, 2).reshape(K, N)
    assert torch.equal(inv, wt)
