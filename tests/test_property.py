"""Property-based tests (hypothesis): dataset index builders and the
cosine_restarts schedule hold their invariants on arbitrary inputs."""

import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

from relora_amd.data.gpt2_dataset import _num_epochs, build_sample_idx_py
from relora_amd.data.megatron import get_train_valid_test_split_
from relora_amd.training_utils import get_scheculer


@settings(max_examples=40, deadline=None, derandomize=True)
@given(
    sizes=st.lists(st.integers(1, 40), min_size=2, max_size=40),
    seq_length=st.integers(2, 24),
    num_epochs=st.integers(1, 3),
    seed=st.integers(0, 2**31 - 1),
)
def test_sample_idx_windows_are_exact(sizes, seq_length, num_epochs, seed):
    """Every sample addressed by sample_idx spans exactly seq_length+1 tokens
    of the shuffled document stream, and consecutive samples overlap by one."""
    sizes = np.asarray(sizes, dtype=np.int32)
    rng = np.random.RandomState(seed)
    doc_idx = np.tile(np.arange(len(sizes), dtype=np.int32), num_epochs)
    rng.shuffle(doc_idx)
    tokens_per_epoch = int(sizes.sum())
    sample_idx = build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs,
                                     tokens_per_epoch)
    # cumulative token position of (doc_cursor, offset)
    csum = np.concatenate([[0], np.cumsum(sizes[doc_idx])])

    def pos(row):
        d, off = sample_idx[row]
        return csum[d] + off

    n = sample_idx.shape[0] - 1
    assert n == (num_epochs * tokens_per_epoch - 1) // seq_length
    for i in range(n):
        # window i covers [pos(i), pos(i+1)] inclusive = seq_length+1 tokens
        assert pos(i + 1) - pos(i) == seq_length
    # C++ builder agrees when available
    try:
        from relora_amd.data import _index_helpers as helpers
    except ImportError:
        return
    cpp = helpers.build_sample_idx_int32(sizes, doc_idx, seq_length, num_epochs,
                                         tokens_per_epoch)
    np.testing.assert_array_equal(cpp, sample_idx)


@settings(max_examples=30, deadline=None, derandomize=True)
@given(
    weights=st.lists(st.floats(0.05, 10.0), min_size=1, max_size=8),
    size=st.integers(1, 400),
)
def test_blending_indices_invariants(weights, size):
    """Per-dataset sample indices are sequential 0..k-1 and the split tracks
    the weights within one sample per dataset."""
    helpers = pytest.importorskip("relora_amd.data._index_helpers")
    w = np.asarray(weights, dtype=np.float64)
    w = w / w.sum()
    di = np.zeros(size, dtype=np.uint8)
    dsi = np.zeros(size, dtype=np.int64)
    helpers.build_blending_indices(di, dsi, w, len(w), size, False)
    for d in range(len(w)):
        mine = dsi[di == d]
        np.testing.assert_array_equal(mine, np.arange(len(mine)))
    counts = np.bincount(di, minlength=len(w)).astype(float)
    # greedy largest-deficit keeps |count - w*size| <= 1 + w*1 slack
    assert np.all(np.abs(counts - w * size) <= 2 + w * 2)


@settings(max_examples=25, deadline=None, derandomize=True)
@given(
    restart_every=st.integers(2, 50),
    cycles=st.integers(1, 6),
    warmup=st.integers(1, 20),
    restart_warmup=st.integers(1, 10),
    min_lr_ratio=st.floats(0.0, 0.9),
)
def test_cosine_restarts_schedule_invariants(restart_every, cycles, warmup,
                                             restart_warmup, min_lr_ratio):
    total = restart_every * cycles
    if warmup >= restart_every:
        warmup = restart_every - 1
    if warmup == 0:
        warmup = 1
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=1.0)
    try:
        sched = get_scheculer(
            optimizer=opt, scheduler_type="cosine_restarts",
            num_training_steps=total, warmup_steps=warmup,
            min_lr_ratio=min_lr_ratio, cycle_length=None,
            restart_warmup_steps=restart_warmup, adjust_step=0,
        )
    except ValueError:
        return  # invalid combos are allowed to be rejected
    lrs = []
    for _ in range(total):
        lrs.append(opt.param_groups[0]["lr"])
        opt.step()
        sched.step()
    lrs = np.asarray(lrs)
    assert np.all(lrs >= -1e-9) and np.all(lrs <= 1.0 + 1e-9)
    # warmup is non-decreasing
    assert np.all(np.diff(lrs[:warmup]) >= -1e-9)
    # at each restart boundary (>= second cycle), the step right after the
    # boundary begins a re-warmup: lr climbs over the restart_warmup window
    for c in range(1, cycles):
        b = c * restart_every
        seg = lrs[b:min(b + restart_warmup, total)]
        if len(seg) >= 2:
            assert np.all(np.diff(seg) >= -1e-6), (c, seg)


@settings(max_examples=25, deadline=None, derandomize=True)
@given(
    docs=st.lists(st.lists(st.integers(0, 60000), min_size=1, max_size=50),
                  min_size=1, max_size=20),
    vocab=st.sampled_from([100, 40000, 70000]),
)
def test_mmap_indexed_dataset_roundtrip_and_slices(tmp_path_factory, docs, vocab):
    """MMapIndexedDataset: arbitrary corpora round-trip bit-exactly and
    get(idx, offset, length) windows match the source (the contract
    GPT2Dataset sample addressing relies on)."""
    import numpy as np

    from relora_amd.data import indexed_dataset as idx

    tmp = tmp_path_factory.mktemp("mmapprop")
    prefix = str(tmp / "corpus")
    dtype = idx.best_fitting_dtype(vocab)
    docs = [[t % vocab for t in d] for d in docs]
    builder = idx.make_builder(prefix + ".bin", "mmap", vocab_size=vocab)
    for d in docs:
        builder.add_item(torch.tensor(d, dtype=torch.int64))
        builder.end_document()
    builder.finalize(prefix + ".idx")

    ds = idx.make_dataset(prefix, "mmap", skip_warmup=True)
    assert len(ds) == len(docs)
    assert ds[0].dtype == np.dtype(dtype)
    for i, d in enumerate(docs):
        np.testing.assert_array_equal(ds[i], np.asarray(d))
        # offset/length windows
        if len(d) > 1:
            win = ds.get(i, offset=1, length=len(d) - 1)
            np.testing.assert_array_equal(win, np.asarray(d[1:]))
    np.testing.assert_array_equal(ds.sizes, [len(d) for d in docs])
