"""Paged KV manager invariants (mirrors reference tests/test_paged_kv.py and
test_cache.py: alloc semantics, budget backpressure, commit/rollback)."""
import threading
import time

import pytest
import torch

from bloombee_amd.kv import AllocationFailed, PagedKVCache
from bloombee_amd.ops import reference as ref


def make_cache(max_tokens=1024, layers=2, heads=2, dim=16, page=16):
    return PagedKVCache(layers, heads, dim, page_size=page, max_tokens=max_tokens,
                        device="cpu", dtype=torch.float32)


def test_alloc_and_release_budget():
    c = make_cache(max_tokens=1024)
    assert c.tokens_left == 1024
    h = c.allocate(batch_size=2, max_length=256)
    assert c.tokens_left == 1024 - 512
    h.close()
    assert c.tokens_left == 1024


def test_alloc_too_large_raises():
    c = make_cache(max_tokens=128)
    with pytest.raises(AllocationFailed):
        c.allocate(batch_size=2, max_length=128)


def test_alloc_timeout_then_unblock():
    c = make_cache(max_tokens=256)
    h1 = c.allocate(2, 64)
    with pytest.raises(AllocationFailed):
        c.allocate(2, 96, timeout=0.05)

    results = {}

    def waiter():
        results["h"] = c.allocate(2, 96, timeout=5.0)

    t = threading.Thread(target=waiter)
    t.start()
    time.sleep(0.05)
    h1.close()
    t.join(timeout=5)
    assert "h" in results
    results["h"].close()


def test_extend_allocates_pages_lazily():
    c = make_cache(max_tokens=1024)
    h = c.allocate(1, 256)
    assert len(h.seqs[0].pages) == 0
    h.extend(17)
    assert len(h.seqs[0].pages) == 2  # 17 tokens -> 2 pages of 16
    assert h.lengths == [17]
    h.close()


def test_commit_rollback_frees_pages():
    c = make_cache(max_tokens=1024)
    h = c.allocate(1, 256)
    h.extend(16)                      # committed
    free_before = len(c._free_pages)
    h.extend(40, speculative=True)    # spec tokens -> 3 extra pages (56 tok)
    assert h.lengths == [56] and h.committed_lengths == [16]
    h.commit(accepted=[8])            # keep 8 of 40
    assert h.committed_lengths == [24]
    assert h.lengths == [24]
    assert len(c._free_pages) == free_before - 1  # 24 tokens -> 2 pages
    h.rollback()                      # no-op now
    assert h.lengths == [24]
    h.close()


def test_truncate_failover():
    c = make_cache(max_tokens=1024)
    h = c.allocate(1, 256)
    h.extend(100)
    h.truncate([30])
    assert h.committed_lengths == [30]
    assert len(h.seqs[0].pages) == 2
    h.close()


def test_write_gather_roundtrip():
    torch.manual_seed(0)
    c = make_cache(max_tokens=1024, layers=1, heads=2, dim=16)
    h = c.allocate(2, 64)
    T = 20
    start = torch.tensor([s.l_spec for s in h.seqs], dtype=torch.int32)
    h.extend(T)
    k = torch.randn(2, 2, T, 16)
    v = torch.randn(2, 2, T, 16)
    ref.kv_write(k, v, h.k_pages(0), h.v_pages(0), h.page_table(), start)
    for b in range(2):
        kg, vg = ref.kv_gather(h.k_pages(0), h.v_pages(0), h.page_table(), T, b)
        assert torch.equal(kg, k[b])
        assert torch.equal(vg, v[b])
    # append one more token at position T
    start2 = torch.tensor([T, T], dtype=torch.int32)
    h.extend(1)
    k2 = torch.randn(2, 2, 1, 16)
    v2 = torch.randn(2, 2, 1, 16)
    ref.kv_write(k2, v2, h.k_pages(0), h.v_pages(0), h.page_table(), start2)
    kg, vg = ref.kv_gather(h.k_pages(0), h.v_pages(0), h.page_table(), T + 1, 0)
    assert torch.equal(kg[:, -1], k2[0, :, 0])
    h.close()


def test_independent_page_lists_per_sequence():
    c = make_cache(max_tokens=1024)
    h = c.allocate(3, 128)
    h.extend(33)
    pages = [set(s.pages) for s in h.seqs]
    assert pages[0] & pages[1] == set()
    assert pages[1] & pages[2] == set()
    h.close()


def test_session_swap_out_in_roundtrip():
    """KV host offload for session multiplexing (ref micro-batch KV
    offload/prefetch): swap out -> pages reusable by others -> swap in ->
    identical contents at (possibly) different physical pages."""
    import torch

    from bloombee_amd.kv.paged import PagedKVCache
    from bloombee_amd.ops import reference as ref

    pool = PagedKVCache(num_layers=2, num_kv_heads=2, head_dim=16,
                        page_size=4, max_tokens=64)
    h1 = pool.allocate(1, 16)
    k = torch.randn(1, 2, 10, 16).to(torch.bfloat16)
    v = torch.randn(1, 2, 10, 16).to(torch.bfloat16)
    h1.extend(10)
    for l in range(2):
        ref.kv_write(k, v, h1.k_pages(l), h1.v_pages(l), h1.page_table(),
                     torch.zeros(1, dtype=torch.int32))
    before = [ref.kv_gather(h1.k_pages(l), h1.v_pages(l), h1.page_table(), 10, 0)
              for l in range(2)]
    old_pages = list(h1.seqs[0].pages)

    h1.swap_out()
    assert h1.is_swapped
    # freed pages are usable by another session
    h2 = pool.allocate(1, 16)
    h2.extend(12)
    k2 = torch.randn(1, 2, 12, 16).to(torch.bfloat16)
    for l in range(2):
        ref.kv_write(k2, k2, h2.k_pages(l), h2.v_pages(l), h2.page_table(),
                     torch.zeros(1, dtype=torch.int32))

    h1.swap_in()
    assert not h1.is_swapped
    after = [ref.kv_gather(h1.k_pages(l), h1.v_pages(l), h1.page_table(), 10, 0)
             for l in range(2)]
    for (kb, vb), (ka, va) in zip(before, after):
        assert torch.equal(kb, ka)
        assert torch.equal(vb, va)
    # decode continues: extend past the swap
    h1.extend(2)
    h2.close()
    h1.close()


def test_swap_out_to_disk_roundtrip(tmp_path):
    """Disk KV tier (ref TorchMixedDevice GPU/CPU/disk partition): a session
    swapped to a file restores byte-identical KV into fresh pages."""
    import torch

    from bloombee_amd.kv.paged import PagedKVCache

    cache = PagedKVCache(num_layers=2, num_kv_heads=2, head_dim=16,
                         max_tokens=1 << 10, device="cpu",
                         dtype=torch.float32)
    h = cache.allocate(2, 128)
    h.extend(40)
    for l in range(2):
        cache.k_pages(l).normal_()
        cache.v_pages(l).normal_()
    before = [(cache.k_pages(l)[torch.tensor(h.seqs[0].pages)].clone(),
               cache.v_pages(l)[torch.tensor(h.seqs[0].pages)].clone())
              for l in range(2)]
    h.swap_out(to_disk=True, disk_dir=str(tmp_path))
    assert h.is_swapped
    assert any(p.suffix == ".pt" for p in tmp_path.iterdir())
    # scribble over the whole pool while the session is on disk
    for l in range(2):
        cache.k_pages(l).zero_()
        cache.v_pages(l).zero_()
    h.swap_in()
    assert not h.is_swapped
    assert not any(p.suffix == ".pt" for p in tmp_path.iterdir())
    for l in range(2):
        k = cache.k_pages(l)[torch.tensor(h.seqs[0].pages)]
        v = cache.v_pages(l)[torch.tensor(h.seqs[0].pages)]
        assert torch.equal(k, before[l][0])
        assert torch.equal(v, before[l][1])
    h.close()


def test_swap_out_compressed_roundtrip(tmp_path):
    """compress_cache tier: 4-bit quantized swap snapshots restore within the
    group-quant tolerance (ref flexgen cache compression)."""
    import torch

    from bloombee_amd.kv.paged import PagedKVCache

    cache = PagedKVCache(num_layers=1, num_kv_heads=2, head_dim=32,
                         max_tokens=1 << 10, device="cpu",
                         dtype=torch.float32)
    h = cache.allocate(1, 128)
    h.extend(48)
    cache.k_pages(0).uniform_(-1, 1)
    cache.v_pages(0).uniform_(-1, 1)
    pages = torch.tensor(h.seqs[0].pages)
    k0 = cache.k_pages(0)[pages].clone()
    v0 = cache.v_pages(0)[pages].clone()
    h.swap_out(to_disk=True, disk_dir=str(tmp_path), compress=True)
    # compressed file must be much smaller than the raw snapshot
    f = next(p for p in tmp_path.iterdir() if p.suffix == ".pt")
    raw_bytes = (k0.numel() + v0.numel()) * 4
    assert f.stat().st_size < raw_bytes * 0.5, (f.stat().st_size, raw_bytes)
    cache.k_pages(0).zero_()
    cache.v_pages(0).zero_()
    h.swap_in()
    k1 = cache.k_pages(0)[torch.tensor(h.seqs[0].pages)]
    v1 = cache.v_pages(0)[torch.tensor(h.seqs[0].pages)]
    assert (k1 - k0).abs().max() < 0.07  # 4-bit over [-1,1]: step ~ 2/15
    assert (v1 - v0).abs().max() < 0.07
    h.close()


def test_swap_out_close_no_double_free():
    """Closing a swapped-out session must not put its (already freed) page
    ids back in the free list — duplicates would let two sessions share a
    physical page."""
    c = make_cache(max_tokens=128, page=16)  # 8-page pool
    h = c.allocate(1, 64)
    h.extend(33)  # 3 pages
    h.swap_out()
    assert sorted(c._free_pages) == list(range(c.n_pages))
    h.close()  # must NOT re-free the swapped pages
    assert sorted(c._free_pages) == list(range(c.n_pages))
    assert len(c._free_pages) == c.n_pages


def test_swap_out_rollback_truncate_then_swap_in():
    c = make_cache(max_tokens=256, page=16)
    h = c.allocate(1, 128)
    h.extend(40)
    k0 = h.k_pages(0)[int(h.page_table_host()[0, 0]), 0, 3, :].clone()
    h.swap_out()
    # truncate while swapped: no pages held, must not free anything twice
    h.truncate([20])
    free_before = sorted(c._free_pages)
    assert len(free_before) == len(set(free_before))
    h.swap_in()
    # only ceil(20/16)=2 pages retained after the trim
    assert len(h.seqs[0].pages) == 2
    assert h.lengths == [20]
    k1 = h.k_pages(0)[int(h.page_table_host()[0, 0]), 0, 3, :]
    assert torch.equal(k0, k1)
    free_now = c._free_pages
    assert len(free_now) == len(set(free_now))
    h.close()
    assert sorted(c._free_pages) == list(range(c.n_pages))


def test_swap_out_extend_guarded():
    from bloombee_amd.kv.paged import PagedKVError
    c = make_cache(max_tokens=128)
    h = c.allocate(1, 64)
    h.extend(16)
    h.swap_out()
    with pytest.raises(PagedKVError):
        h.extend(1)
    h.close()


def test_row_swap_roundtrip_and_extend_rows():
    """Row-granular staging (micro-batch KV multiplexing): swapped rows
    advance length-only; swap_in restores data into fresh pages; resident
    rows are unaffected."""
    c = make_cache(max_tokens=512, page=16)
    h = c.allocate(4, 128)
    h.extend(20)
    # write recognizable data into row 2's pages
    marker = torch.full((2, 16, 16), 7.0)
    for l in range(2):
        for j, pg in enumerate(h.seqs[2].pages):
            c.k_pages(l)[pg] = float(l * 10 + j)
    h.swap_out_rows(2, 4)
    assert h.rows_swapped(2, 4) and not h.rows_swapped(0, 2)
    assert h.seqs[2].pages == [] and h.seqs[3].pages == []
    free_now = c._free_pages
    assert len(free_now) == len(set(free_now))
    # grow ALL rows while 2..4 are swapped: only resident rows take pages
    h.extend(13)  # 20 -> 33 tokens: 3 pages per seq
    assert len(h.seqs[0].pages) == 3 and len(h.seqs[2].pages) == 0
    assert h.lengths == [33, 33, 33, 33]
    h.swap_in_rows(2, 4)
    assert not h.rows_swapped(2, 4)
    assert len(h.seqs[2].pages) == 3  # grown while away
    for l in range(2):
        for j in range(2):  # snapshot covered the first 2 pages
            pg = h.seqs[2].pages[j]
            assert torch.all(c.k_pages(l)[pg] == float(l * 10 + j)), (l, j)
    # per-row extends (micro-batch slices grow only their own rows)
    h.extend_rows(0, 2, 16)
    assert h.lengths == [49, 49, 33, 33]
    h.extend_rows(2, 4, 16)
    assert h.lengths == [49, 49, 49, 49]
    h.close()
    assert sorted(c._free_pages) == list(range(c.n_pages))


def test_row_swap_extend_blocks_until_pages_freed():
    """extend_rows under page pressure waits (without deadlocking the
    staging thread) until swap_out_rows frees pages."""
    c = make_cache(max_tokens=64, page=16)  # 4-page pool
    h = c.allocate(4, 64, resident_batch=1)
    h.extend_rows(0, 2, 32)  # rows 0,1 take all 4 pages
    import threading

    def stage():
        time.sleep(0.05)
        h.swap_out_rows(0, 2)

    t = threading.Thread(target=stage)
    t.start()
    h.extend_rows(2, 4, 32, timeout=5)  # blocks until the swap frees pages
    t.join()
    assert len(h.seqs[2].pages) == 2 and h.seqs[0].pages == []
    assert h.lengths == [32, 32, 32, 32]
    h.close()


def test_swap_auto_policy_quantities():
    """mixed_attn="auto" decides restore-vs-mixed from swapped_pages_needed()
    vs free_page_count(): after swap-out the pool reports the freed pages,
    and page scarcity flips the decision to mixed-device decode."""
    from bloombee_amd.kv.paged import PagedKVCache

    cache = PagedKVCache(num_layers=1, num_kv_heads=2, head_dim=64,
                         max_tokens=16 * 8, page_size=16, device="cpu",
                         dtype=torch.bfloat16)
    h = cache.allocate(2, 16 * 2)   # 2 seqs x up to 2 pages each
    h.extend(20)                    # 2 pages per seq -> 4 pages used
    used = 4
    assert cache.free_page_count() == 8 - used
    h.swap_out()
    assert h.swapped_pages_needed() == used
    assert cache.free_page_count() == 8   # pages returned to the pool
    # fits -> auto restores
    assert h.swapped_pages_needed() <= cache.free_page_count()
    # page scarcity (e.g. other resident sessions) -> auto goes mixed
    held = cache._take_pages(6)
    assert cache.free_page_count() == 2
    assert h.swapped_pages_needed() > cache.free_page_count()
    cache._give_pages(held)
    h.swap_in()
    assert not h.is_swapped and h.swapped_pages_needed() == 0
    h.close()
