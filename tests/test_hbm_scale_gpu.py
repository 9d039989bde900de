"""288 GB HBM sizing: the sparse-embedding PS holds large resident
tables per GPU (fp32 masters + bf16 shadow) and serves gather/
scatter-add through the HIP kernels at HBM rates."""

import time

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(420)
def test_large_embedding_table_push_pull():
    from tfmesos_amd.ps.sparse import EmbeddingTable

    # 24M x 256 fp32 master (24.6 GB) + bf16 shadow (12.3 GB) ~= 37 GB
    # resident — an eighth-scale slice of one MI355X's 288 GB
    V, D = 24_000_000, 256
    t = EmbeddingTable("big", rows=V, dim=D, device="cuda:0", lr=0.1)
    torch.cuda.synchronize()
    ids = torch.randint(0, V, (1_000_000,), device="cuda:0")

    rows = t.pull(ids)                      # HIP gather, 1M x 256 bf16
    assert rows.shape == (1_000_000, D) and rows.dtype == torch.bfloat16
    ref = t.master[ids[:1000]].to(torch.bfloat16)
    assert torch.equal(rows[:1000], ref)

    grads = torch.randn(1_000_000, D, device="cuda:0", dtype=torch.bfloat16)
    before = t.master[ids[:8]].clone()
    t.push(ids, grads, lr=0.5)              # HIP scatter-add + shadow refresh
    after = t.master[ids[:8]]
    assert not torch.equal(before, after)
    torch.cuda.synchronize()

    # throughput sanity: pulls move 0.5 GB each; expect well over 100 GB/s
    n = 20
    t0 = time.perf_counter()
    for _ in range(n):
        t.pull(ids)
    torch.cuda.synchronize()
    gbps = n * ids.numel() * D * 2 / (time.perf_counter() - t0) / 1e9
    print("gather throughput: %.0f GB/s" % gbps)
    assert gbps > 100, gbps


@pytest.mark.timeout(420)
def test_large_flat_dense_store_apply():
    """Dense PS sizing for 288 GB HBM: a 10B-parameter flat store
    (40 GB fp32 masters + 20 GB bf16 shadow + 40 GB fp32 grads ~= 100 GB
    resident) updated by ONE fused apply kernel at HBM rates."""
    from tfmesos_amd import ops

    n = 10_000_000_000
    p = torch.zeros(n, dtype=torch.float32, device="cuda:0")
    shadow = torch.zeros(n, dtype=torch.bfloat16, device="cuda:0")
    g = torch.ones(n, dtype=torch.float32, device="cuda:0")
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    ops.fused_sgd(p, g, lr=0.5, bf16_out=shadow)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # traffic: read p + g, write p + shadow = 14 bytes/param
    gbps = 14.0 * n / dt / 1e9
    print("fused apply on 10B params: %.3f s, %.0f GB/s" % (dt, gbps))
    assert float(p[0]) == -0.5 and float(p[-1]) == -0.5
    assert float(shadow[n // 2].float()) == -0.5
    assert gbps > 1000, gbps
