"""Batched-CFG substrate tests: CSR/CSC vs dense adjacency oracle."""

import torch

from deepdfa_amd.graph import BatchedCFG, batch_graphs, synthetic_cfg_batch
from deepdfa_amd.graph.synthetic import synthetic_cfg


def dense_adj(g: BatchedCFG) -> torch.Tensor:
    """A[v][u] = count of edges u->v, from the in-CSR."""
    N = g.num_nodes
    A = torch.zeros(N, N)
    ip = g.indptr.tolist()
    idx = g.indices.tolist()
    for v in range(N):
        for e in range(ip[v], ip[v + 1]):
            A[v, idx[e]] += 1
    return A


def test_from_edges_self_loops():
    g = BatchedCFG.from_edges(4, [0, 1, 2], [1, 2, 3], add_self_loops=True)
    assert g.num_nodes == 4
    assert g.num_edges == 3 + 4
    A = dense_adj(g)
    assert A.diagonal().sum() == 4
    assert A[1, 0] == 1 and A[2, 1] == 1 and A[3, 2] == 1


def test_duplicate_self_loop_not_doubled():
    g = BatchedCFG.from_edges(2, [0, 0], [0, 1], add_self_loops=True)
    A = dense_adj(g)
    assert A[0, 0] == 1.0  # existing self-loop removed then re-added once


def test_csc_is_transpose():
    gen = torch.Generator().manual_seed(7)
    g = synthetic_cfg(gen)
    A = dense_adj(g)
    # build dense from CSC
    N = g.num_nodes
    At = torch.zeros(N, N)
    tp = g.t_indptr.tolist()
    tidx = g.t_indices.tolist()
    for u in range(N):
        for e in range(tp[u], tp[u + 1]):
            At[u, tidx[e]] += 1
    assert torch.equal(A, At.t())


def test_batching_block_diagonal():
    gen = torch.Generator().manual_seed(0)
    gs = [synthetic_cfg(gen) for _ in range(3)]
    gb = batch_graphs(gs)
    assert gb.num_graphs == 3
    assert gb.num_nodes == sum(g.num_nodes for g in gs)
    assert gb.num_edges == sum(g.num_edges for g in gs)
    Ab = dense_adj(gb)
    off = 0
    for g in gs:
        n = g.num_nodes
        assert torch.equal(Ab[off : off + n, off : off + n], dense_adj(g))
        # off-diagonal blocks are zero
        Ab[off : off + n, off : off + n] = 0
        off += n
    assert Ab.sum() == 0


def test_unbatch_roundtrip():
    g = synthetic_cfg_batch(4, seed=3)
    parts = g.unbatch()
    g2 = batch_graphs(parts)
    assert torch.equal(g.indptr, g2.indptr)
    assert torch.equal(g.indices, g2.indices)
    assert torch.equal(g.ndata["_VULN"], g2.ndata["_VULN"])


def test_save_load_roundtrip(tmp_path):
    g = synthetic_cfg_batch(2, seed=5)
    p = str(tmp_path / "g.pt")
    g.save(p)
    g2 = BatchedCFG.load(p)
    assert torch.equal(g.indices, g2.indices)
    assert set(g.ndata) == set(g2.ndata)


def test_segment_ids():
    g = synthetic_cfg_batch(3, seed=1)
    seg = g.segment_ids()
    counts = torch.bincount(seg.to(torch.int64), minlength=3)
    assert torch.equal(counts, g.batch_num_nodes())


def test_synthetic_batch_deterministic():
    a = synthetic_cfg_batch(8, seed=42)
    b = synthetic_cfg_batch(8, seed=42)
    assert torch.equal(a.indices, b.indices)
    assert torch.equal(a.ndata["_ABS_DATAFLOW_api"], b.ndata["_ABS_DATAFLOW_api"])
    c = synthetic_cfg_batch(8, seed=43)
    assert not torch.equal(a.ndata["_ABS_DATAFLOW_api"], c.ndata["_ABS_DATAFLOW_api"])


def test_pad_batch_fuzz_invariants():
    """Property fuzz over random batches x adversarial quanta: pad_batch
    must always produce exactly the bucket shape, a monotone CSR whose
    in/out edge counts agree, zeroed dummy features, and a weight mask
    selecting exactly the real graphs."""
    import torch

    from deepdfa_amd.graph.pad import bucket_shape, pad_batch
    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch

    for seed in range(12):
        bs = 3 + (seed * 7) % 40
        g = synthetic_cfg_batch(bs, seed=seed)
        node_q = [64, 256, 1024, 4096][seed % 4]
        edge_q = [32, 64, 512, 4096][(seed + 1) % 4]
        shape = bucket_shape(g, b_pad=bs + 1 + seed % 3, node_q=node_q, edge_q=edge_q)
        padded, w = pad_batch(g, *shape)
        assert (padded.num_nodes, padded.num_edges, padded.num_graphs) == shape
        assert int(w.sum()) == bs and w.numel() == shape[2]
        d = padded.indptr.to(torch.int64).diff()
        assert (d >= 0).all() and int(padded.indptr[-1]) == padded.num_edges
        assert int(padded.t_indptr[-1]) == padded.num_edges
        assert (padded.indices < padded.num_nodes).all()
        # real prefix untouched; dummy features zero
        assert torch.equal(padded.indices[: g.num_edges], g.indices)
        for k, v in padded.ndata.items():
            assert (v[g.num_nodes:] == 0).all(), k
        # per-graph node offsets cover [0, n_pad] monotonically
        off = padded.node_offsets.to(torch.int64)
        assert int(off[0]) == 0 and int(off[-1]) == padded.num_nodes
        assert (off.diff() >= 0).all()
