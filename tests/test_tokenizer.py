"""Tokenizers: CLIP-BPE layout, roundtrip, padding semantics."""

import pytest
import torch

from dalle_pytorch_amd.utils.tokenizer import (
    SimpleTokenizer, byte_unicode_table, default_bpe_path)


def test_byte_table_reversible():
    t = byte_unicode_table()
    assert len(t) == 256 and len(set(t.values())) == 256


def test_vocab_layout_fixed():
    tok = SimpleTokenizer()
    assert tok.vocab_size == 49408
    assert tok.encoder['<|startoftext|>'] == 49406
    assert tok.encoder['<|endoftext|>'] == 49407


def test_encode_decode_roundtrip():
    tok = SimpleTokenizer()
    text = 'a red square on a blue background'
    ids = tok.encode(text)
    assert len(ids) > 0
    assert tok.decode(ids).strip() == text


def test_tokenize_padding_and_truncate():
    tok = SimpleTokenizer()
    out = tok.tokenize(['hello world', 'hi'], context_length=16)
    assert out.shape == (2, 16)
    assert out.dtype == torch.long
    assert (out[1] == 0).sum() > (out[0] == 0).sum()
    with pytest.raises(RuntimeError):
        tok.tokenize('word ' * 300, context_length=8)
    t = tok.tokenize('word ' * 300, context_length=8, truncate_text=True)
    assert t.shape == (1, 8)


@pytest.mark.skipif(default_bpe_path() is None, reason='no merges data file')
def test_bpe_merges_active():
    tok = SimpleTokenizer()
    assert len(tok.bpe_ranks) > 1000
    # a common word must merge to fewer tokens than its letters
    ids = tok.encode('hello')
    assert len(ids) < 5


def test_decode_skips_pad_tokens():
    tok = SimpleTokenizer()
    ids = tok.encode('cat')
    pad = set(range(49408 - 256, 49408))
    text = tok.decode(list(ids) + [49408 - 10], pad_tokens=pad)
    assert 'cat' in text
