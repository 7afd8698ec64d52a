"""Property-based invariant tests (hypothesis) for the core data
structures — the tier the reference lacks (its paged-table
invariants are example-based; here random operation sequences check the
same contracts exhaustively)."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from bloombee_amd.data_structures import (RemoteModuleInfo, ServerInfo,
                                           compute_spans)
from bloombee_amd.kv.paged import PagedKVCache
from bloombee_amd.net.tensors import pack_frame, unpack_frame
from bloombee_amd.spec.tree import TokenTree

# ---------------------------------------------------------------------------
# Paged KV: random extend/commit/rollback/truncate/swap sequences must
# conserve the page budget and keep per-seq page counts consistent.
# ---------------------------------------------------------------------------

op_strategy = st.lists(
    st.one_of(
        st.tuples(st.just("extend"), st.integers(1, 40),
                  st.booleans()),                       # (n, speculative)
        st.tuples(st.just("commit"), st.just(0), st.just(False)),
        st.tuples(st.just("rollback"), st.just(0), st.just(False)),
        st.tuples(st.just("truncate"), st.integers(0, 64), st.just(False)),
        st.tuples(st.just("swap"), st.just(0), st.just(False)),
    ),
    min_size=1, max_size=30)


@settings(max_examples=60, deadline=None)
@given(ops=op_strategy, batch=st.integers(1, 3))
def test_paged_kv_invariants(ops, batch):
    cache = PagedKVCache(num_layers=1, num_kv_heads=1, head_dim=16,
                         max_tokens=1 << 12, device="cpu",
                         dtype=torch.float32)
    free0 = len(cache._free_pages)
    h = cache.allocate(batch, 256)
    P = cache.page_size
    for op, n, spec in ops:
        try:
            if op == "extend":
                h.extend(n, speculative=spec)
            elif op == "commit":
                h.commit()
            elif op == "rollback":
                h.rollback()
            elif op == "truncate":
                h.truncate([min(n, s.l_acc) for s in h.seqs])
            elif op == "swap":
                h.swap_out()
                h.swap_in()
        except Exception as e:  # only capacity errors are acceptable
            assert "token" in str(e).lower() or "capacity" in str(e).lower(), e
            break
        for s in h.seqs:
            # invariant: l_acc <= l_spec <= pages * P, pages cover l_spec
            assert 0 <= s.l_acc <= s.l_spec
            assert s.l_spec <= len(s.pages) * P
            assert len(set(s.pages)) == len(s.pages)  # no duplicate pages
    h.close()
    # invariant: closing returns every page
    assert len(cache._free_pages) == free0
    assert cache.tokens_left == cache.max_tokens


# ---------------------------------------------------------------------------
# Wire codec: lossless roundtrip for every codec across dtypes/shapes.
# ---------------------------------------------------------------------------

@settings(max_examples=40, deadline=None)
@given(
    shape=st.lists(st.integers(1, 9), min_size=1, max_size=3),
    dtype=st.sampled_from([torch.float32, torch.bfloat16, torch.int32]),
    codec=st.sampled_from(["raw", "zlib", "bsplit+zlib",
                           "bsplit+zlibmt"]),
    seed=st.integers(0, 2 ** 16),
)
def test_wire_codec_roundtrip(shape, dtype, codec, seed):
    gen = torch.Generator().manual_seed(seed)
    if dtype is torch.int32:
        t = torch.randint(-1000, 1000, shape, generator=gen,
                          dtype=torch.int32)
    else:
        t = torch.randn(shape, generator=gen).to(dtype)
    blob = pack_frame({"k": 1}, [t], codec=codec)
    meta, (back,) = unpack_frame(blob)
    assert meta == {"k": 1}
    assert back.dtype == t.dtype and back.shape == t.shape
    assert torch.equal(back, t)  # codecs must be lossless


# ---------------------------------------------------------------------------
# Token tree: the ancestor mask must equal the brute-force parent walk.
# ---------------------------------------------------------------------------

@settings(max_examples=60, deadline=None)
@given(parents=st.lists(st.integers(-5, 30), min_size=1, max_size=24),
       seed=st.integers(0, 999))
def test_token_tree_mask_matches_parent_walk(parents, seed):
    tree = TokenTree()
    for i, p in enumerate(parents):
        parent = p % (i + 1) - 1 if i > 0 else -1  # valid: -1 .. i-1
        tree.add(int(seed + i) % 100, parent, 0.5)
    mask = tree.attention_mask()
    n = len(tree)
    for i in range(n):
        anc = set()
        j = i
        while j != -1:
            anc.add(j)
            j = tree.parents[j]
        for j in range(n):
            assert bool(mask[i, j]) == (j in anc), (i, j, tree.parents)


# ---------------------------------------------------------------------------
# compute_spans: spans must be contiguous, sorted, within each server's range.
# ---------------------------------------------------------------------------

@settings(max_examples=60, deadline=None)
@given(ranges=st.lists(
    st.tuples(st.integers(0, 10), st.integers(1, 6)), min_size=1, max_size=6))
def test_compute_spans_contiguity(ranges):
    # build per-block announcement lists from each server's claimed range
    num_blocks = max(s + n for s, n in ranges)
    infos = [RemoteModuleInfo(uid=f"blk{i}", servers={})
             for i in range(num_blocks)]
    claimed = {}
    for i, (start, length) in enumerate(ranges):
        peer = f"peer{i}"
        claimed[peer] = (start, min(start + length, num_blocks))
        for blk in range(*claimed[peer]):
            infos[blk].servers[peer] = ServerInfo(host="127.0.0.1",
                                                  port=1000 + i)
    spans = compute_spans(infos)
    assert set(spans) == set(claimed)
    for peer, span in spans.items():
        # a span is the (first) maximal contiguous run of the peer's blocks
        assert (span.start, span.end) == claimed[peer]
        assert span.start < span.end


# ---------------------------------------------------------------------------
# Spec-tree accept: reorder_and_commit must compact exactly the kept rows,
# in order, for every (region size, kept subset) — the subtlest KV op.
# ---------------------------------------------------------------------------

@settings(max_examples=80, deadline=None)
@given(
    prefix=st.integers(0, 40),
    spec_n=st.integers(1, 30),
    keep_mask=st.lists(st.booleans(), min_size=30, max_size=30),
    seed=st.integers(0, 99),
)
def test_reorder_and_commit_compacts_kept_rows(prefix, spec_n, keep_mask,
                                               seed):
    cache = PagedKVCache(num_layers=2, num_kv_heads=1, head_dim=16,
                         max_tokens=1 << 10, device="cpu",
                         dtype=torch.float32)
    h = cache.allocate(1, 128)
    if prefix:
        h.extend(prefix)  # committed region
    h.extend(spec_n, speculative=True)
    gen = torch.Generator().manual_seed(seed)
    # stamp every speculative position with a recognizable value
    P = cache.page_size
    stamps = torch.randn(spec_n, 1, 16, generator=gen)
    for j in range(spec_n):
        pos = prefix + j
        pg = h.seqs[0].pages[pos // P]
        for l in range(2):
            cache.k_pages(l)[pg, :, pos % P, :] = stamps[j] + l
            cache.v_pages(l)[pg, :, :, pos % P] = stamps[j] + 10 + l

    kept = [j for j in range(spec_n) if keep_mask[j]]
    h.reorder_and_commit([kept])
    assert h.seqs[0].l_acc == prefix + len(kept)
    assert h.seqs[0].l_spec == prefix + len(kept)
    for d, j in enumerate(kept):
        pos = prefix + d
        pg = h.seqs[0].pages[pos // P]
        for l in range(2):
            assert torch.equal(cache.k_pages(l)[pg, :, pos % P, :],
                               stamps[j] + l), (d, j)
            assert torch.equal(cache.v_pages(l)[pg, :, :, pos % P],
                               stamps[j] + 10 + l), (d, j)
    h.close()


# ---------------------------------------------------------------------------
# Frames: arbitrary msgpack-able metadata + several mixed-dtype tensors must
# survive a frame roundtrip regardless of codec.
# ---------------------------------------------------------------------------

meta_strategy = st.recursive(
    st.one_of(st.integers(-2**40, 2**40), st.text(max_size=12),
              st.booleans(), st.none(),
              st.floats(allow_nan=False, allow_infinity=False)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=8), children, max_size=4)),
    max_leaves=12)


@settings(max_examples=40, deadline=None)
@given(meta=st.dictionaries(st.text(min_size=1, max_size=8), meta_strategy,
                            max_size=4),
       n_tensors=st.integers(0, 3), seed=st.integers(0, 999),
       codec=st.sampled_from(["raw", "bsplit+zlib"]))
def test_frame_roundtrip_meta_and_tensors(meta, n_tensors, seed, codec):
    gen = torch.Generator().manual_seed(seed)
    dts = [torch.float32, torch.bfloat16, torch.int64]
    ts = [torch.randn(3, 5, generator=gen).to(dts[i % 3])
          if dts[i % 3] != torch.int64 else
          torch.randint(0, 9, (3, 5), generator=gen)
          for i in range(n_tensors)]
    blob = pack_frame(meta, ts, codec=codec)
    m2, ts2 = unpack_frame(blob)
    assert m2 == meta
    assert len(ts2) == len(ts)
    for a, b in zip(ts, ts2):
        assert a.dtype == b.dtype and torch.equal(a, b)
