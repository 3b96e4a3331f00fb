"""Property-based tests (hypothesis) for the data layer: pbin roundtrip
across token sizes, continuous-dataset coverage, CharTokenizer roundtrip,
and sampler skip/resume equivalence (reference test strategy §4: the data
path must be byte-exact and resumable)."""

import numpy as np
from hypothesis import given, settings, strategies as st

from modalities_amd.dataloader.packed_data import (EmbeddedStreamData,
                                                   token_size_for_vocab,
                                                   write_pbin)
from modalities_amd.dataloader.samplers import ResumableDistributedSampler
from modalities_amd.tokenization.tokenizer_wrapper import CharTokenizer

DTYPES = {1: np.uint8, 2: np.uint16, 4: np.uint32}


@settings(max_examples=25, deadline=None)
@given(
    token_size=st.sampled_from([1, 2, 4]),
    docs=st.lists(st.integers(min_value=1, max_value=40), min_size=1,
                  max_size=12),
    seed=st.integers(0, 2**31 - 1),
)
def test_pbin_roundtrip_exact(tmp_path_factory, token_size, docs, seed):
    rng = np.random.default_rng(seed)
    hi = 2 ** (8 * token_size) - 1
    arrays = [rng.integers(0, hi, size=n).astype(DTYPES[token_size])
              for n in docs]
    path = tmp_path_factory.mktemp("pbin") / "d.pbin"
    write_pbin(path, arrays, token_size_in_bytes=token_size)
    stream = EmbeddedStreamData(path)
    assert stream.token_size_in_bytes == token_size
    assert len(stream.index_base) == len(arrays)
    assert stream.num_tokens == sum(len(a) for a in arrays)
    for (off, length), a in zip(stream.index_base, arrays):
        np.testing.assert_array_equal(stream.tokens(off, length), a)


@settings(max_examples=30, deadline=None)
@given(text=st.text(
    alphabet=st.characters(min_codepoint=1, max_codepoint=127), max_size=200))
def test_char_tokenizer_ascii_roundtrip(text):
    tok = CharTokenizer()
    ids = tok.tokenize(text)
    assert tok.decode(ids) == text
    assert all(0 <= t < tok.vocab_size for t in ids)


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(4, 200),
    world=st.sampled_from([1, 2, 4]),
    skip_mult=st.integers(0, 6),
    shuffle=st.booleans(),
    seed=st.integers(0, 1000),
)
def test_sampler_skip_equals_tail_of_full_run(n, world, skip_mult, shuffle, seed):
    """Resume contract: skipping k*world global samples yields exactly the
    tail of the unskipped enumeration, per rank."""
    skip = skip_mult * world
    data = list(range(n))
    for rank in range(world):
        full = list(ResumableDistributedSampler(
            dataset=data, rank=rank, num_replicas=world, shuffle=shuffle,
            seed=seed, drop_last=True))
        if skip >= len(full) * world:
            continue
        resumed = list(ResumableDistributedSampler(
            dataset=data, rank=rank, num_replicas=world, shuffle=shuffle,
            seed=seed, drop_last=True, skip_num_global_samples=skip))
        assert resumed == full[skip // world:], (rank, skip)


@settings(max_examples=20, deadline=None)
@given(vocab=st.integers(2, 2**32 - 1))
def test_token_size_for_vocab_bounds(vocab):
    ts = token_size_for_vocab(vocab)
    assert ts in (1, 2, 4)
    assert vocab <= 2 ** (8 * ts)
    if ts > 1:
        smaller = {2: 1, 4: 2}[ts]
        assert vocab > 2 ** (8 * smaller)


@settings(max_examples=50, deadline=None)
@given(
    dp=st.sampled_from([1, 2, 4, 8]),
    mbs=st.integers(1, 8),
    acc=st.integers(1, 8),
    seq=st.sampled_from([128, 1024, 4096]),
    steps=st.integers(1, 10_000),
)
def test_number_conversion_steps_tokens_inverse(dp, mbs, acc, seq, steps):
    """steps -> tokens -> steps must be the identity (warmstart arithmetic
    cannot drift)."""
    from modalities_amd.utils.number_conversion import NumberConversion
    tokens = NumberConversion.get_num_tokens_from_num_steps(
        num_steps=steps, dp_degree=dp, local_micro_batch_size=mbs,
        sequence_length=seq, gradient_accumulation_steps=acc)
    assert tokens == steps * dp * mbs * acc * seq
    back = NumberConversion.get_num_steps_from_num_tokens(
        dp_degree=dp, local_micro_batch_size=mbs, global_num_tokens=tokens,
        sequence_length=seq, gradient_accumulation_steps=acc)
    assert back == steps


@settings(max_examples=30, deadline=None)
@given(
    seen=st.integers(0, 10**12),
    target=st.integers(0, 10**12),
    steps=st.integers(0, 10**6),
    tsteps=st.integers(0, 10**6),
    eid=st.text(alphabet=st.characters(whitelist_categories=("Ll", "Nd")),
                min_size=1, max_size=12),
)
def test_checkpoint_path_regex_roundtrip(tmp_path_factory, seen, target, steps,
                                         tsteps, eid):
    """The checkpoint folder-name schema must round-trip through the
    number-conversion regex family."""
    from pathlib import Path

    from modalities_amd.utils.number_conversion import NumberConversion
    name = (f"eid_{eid}-seen_steps_{steps}-seen_tokens_{seen}"
            f"-target_steps_{tsteps}-target_tokens_{target}")
    p = Path("/ckpt") / name
    assert NumberConversion.get_num_seen_steps_from_checkpoint_path(p) == steps
    assert NumberConversion.get_global_num_seen_tokens_from_checkpoint_path(p) == seen
    assert NumberConversion.get_num_target_steps_from_checkpoint_path(p) == tsteps
    assert NumberConversion.get_global_num_target_tokens_from_checkpoint_path(p) == target


@settings(max_examples=40, deadline=None)
@given(
    seq=st.lists(st.integers(0, 9), min_size=2, max_size=64),
    seed=st.integers(0, 999),
)
def test_loss_masking_matches_naive_scan(seq, seed):
    """The vectorized cumsum span masking must equal a naive per-token scan
    (b=100 opens a span, e=101 closes it; only tokens strictly inside train)."""
    import torch

    from modalities_amd.batch import DatasetBatch
    from modalities_amd.dataloader.dataloader import LossMaskingCollateFnWrapper

    rng = np.random.default_rng(seed)
    tokens = list(seq)
    # sprinkle well-formed marker pairs
    for _ in range(rng.integers(0, 3)):
        i, j = sorted(rng.integers(0, len(tokens) + 1, size=2))
        tokens = tokens[:i] + [100] + tokens[i:j] + [101] + tokens[j:]
    t = torch.tensor([tokens])

    def passthrough(_):
        return DatasetBatch(samples={"input_ids": t.clone()},
                            targets={"target_ids": t.clone()})

    wrapper = LossMaskingCollateFnWrapper(
        passthrough, target_keys_to_mask=["target_ids"],
        loss_ignore_index=-100, b_mask_token_id=100, e_mask_token_id=101)
    got = wrapper(None).targets["target_ids"][0].tolist()

    # naive scan
    expected, depth = [], 0
    for tok in tokens:
        if tok == 100:
            expected.append(-100)
            depth += 1
        elif tok == 101:
            expected.append(-100)
            depth -= 1
        else:
            expected.append(tok if depth > 0 else -100)
    assert got == expected


@settings(max_examples=15, deadline=None)
@given(
    n_docs=st.integers(1, 20),
    seed=st.integers(0, 500),
    token_size=st.sampled_from([1, 2]),
)
def test_shuffle_preserves_document_multiset(tmp_path_factory, n_docs, seed,
                                             token_size):
    """Document-level shuffling must be a permutation: same multiset of
    documents, deterministic for a fixed seed."""
    from modalities_amd.preprocessing.shuffle_data import shuffle_tokenized_data

    rng = np.random.default_rng(seed)
    docs = [rng.integers(0, 2 ** (8 * token_size) - 1,
                         size=int(rng.integers(1, 30))).astype(DTYPES[token_size])
            for _ in range(n_docs)]
    root = tmp_path_factory.mktemp("shuf")
    src = root / "src.pbin"
    write_pbin(src, docs, token_size_in_bytes=token_size)

    out1, out2 = root / "a.pbin", root / "b.pbin"
    shuffle_tokenized_data(src, out1, batch_size=4, seed=seed)
    shuffle_tokenized_data(src, out2, batch_size=4, seed=seed)

    def read_docs(p):
        s = EmbeddedStreamData(p)
        return [tuple(s.tokens(o, l).tolist()) for o, l in s.index_base]

    a, b = read_docs(out1), read_docs(out2)
    assert a == b  # deterministic
    assert sorted(a) == sorted(tuple(d.tolist()) for d in docs)  # permutation


@settings(max_examples=30, deadline=None)
@given(
    T=st.integers(2, 64).filter(lambda t: t % 2 == 0),
    cp=st.sampled_from([1, 2, 4]),
    B=st.integers(1, 3),
)
def test_cp_target_slices_partition_sequence(T, cp, B):
    """CP target slices across ranks must tile the sequence exactly."""
    import torch

    from modalities_amd.parallel.cp import slice_targets_for_cp
    if T % cp != 0:
        return
    t = torch.arange(B * T).view(B, T)
    parts = [slice_targets_for_cp(t, r, cp) for r in range(cp)]
    assert all(p.shape == (B, T // cp) for p in parts)
    recon = torch.cat(parts, dim=1)
    assert torch.equal(recon, t)
