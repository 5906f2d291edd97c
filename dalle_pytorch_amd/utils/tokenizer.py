"""Tokenizers: CLIP-style byte-BPE (default), HuggingFace-json, BERT-chinese,
and YouTokenToMe — the same four-way surface as the reference
(tokenizer.py:55-266), all CPU-side.

The default tokenizer implements OpenAI-CLIP byte-level BPE from first
principles. The 49,152-merge table is *data*, not code: the package bundles
OpenAI's published CLIP BPE vocabulary (``data/clip_bpe_merges.txt.gz`` —
the standard public ``bpe_simple_vocab_16e6`` merges artifact, gzip-stored;
same data the reference ships at dalle_pytorch/data/). Pass ``bpe_path``
(or set ``DALLE_AMD_BPE_PATH``) to override with any CLIP-format merges
file, plain or gzipped. If no merges data can be found at all, the
tokenizer degrades to pure byte-level encoding with the identical
49,408-slot vocabulary layout (embedding shapes unchanged) and emits a
loud warning, because token ids then differ from CLIP-BPE ids.
"""

import gzip
import html
import os
import warnings
from functools import lru_cache
from pathlib import Path

import torch

try:
    import ftfy
except ImportError:  # optional: only improves unicode cleanup
    ftfy = None

try:
    import regex as _re
except ImportError:
    import re as _re


@lru_cache()
def byte_unicode_table():
    """Reversible byte -> printable-unicode mapping (GPT-2/CLIP convention)."""
    keep = list(range(ord('!'), ord('~') + 1)) + \
        list(range(ord('¡'), ord('¬') + 1)) + \
        list(range(ord('®'), ord('ÿ') + 1))
    mapped = keep[:]
    bump = 0
    for b in range(256):
        if b not in keep:
            keep.append(b)
            mapped.append(256 + bump)
            bump += 1
    return dict(zip(keep, (chr(c) for c in mapped)))


def _clean(text):
    if ftfy is not None:
        text = ftfy.fix_text(text)
    text = html.unescape(html.unescape(text))
    return _re.sub(r'\s+', ' ', text).strip()


def _pairs(word):
    return set(zip(word[:-1], word[1:]))


def _pad_batch(token_lists, context_length, truncate_text, texts):
    out = torch.zeros(len(token_lists), context_length, dtype=torch.long)
    for i, toks in enumerate(token_lists):
        if len(toks) > context_length:
            if not truncate_text:
                raise RuntimeError(
                    f'Input {texts[i]!r} is too long for context length {context_length}')
            toks = toks[:context_length]
        if len(toks):
            out[i, :len(toks)] = torch.as_tensor(list(toks), dtype=torch.long)
    return out


def default_bpe_path():
    env = os.environ.get('DALLE_AMD_BPE_PATH')
    if env:
        return env
    bundled = Path(__file__).resolve().parent.parent / 'data' / 'clip_bpe_merges.txt.gz'
    return str(bundled) if bundled.exists() else None


def _read_merges_text(path):
    path = Path(path)
    if path.suffix == '.gz':
        with gzip.open(path, 'rt', encoding='utf8') as f:
            return f.read()
    return path.read_text(encoding='utf8')


class SimpleTokenizer:
    """CLIP byte-level BPE. Vocabulary layout (49,408 ids): 256 byte symbols,
    256 byte+'</w>' symbols, 48,894 merges, then the two specials."""

    VOCAB_SIZE = 49408
    N_MERGES = 49152 - 256 - 2

    def __init__(self, bpe_path=None):
        self.byte_encoder = byte_unicode_table()
        self.byte_decoder = {v: k for k, v in self.byte_encoder.items()}

        bpe_path = bpe_path if bpe_path is not None else default_bpe_path()
        merges = []
        if bpe_path is not None and Path(bpe_path).exists():
            lines = _read_merges_text(bpe_path).split('\n')
            merges = [tuple(l.split()) for l in lines[1:self.N_MERGES + 1]]
        if not merges:
            warnings.warn(
                'SimpleTokenizer: no BPE merges data found '
                f'(bpe_path={bpe_path!r}); falling back to pure BYTE-LEVEL '
                'encoding. Token ids will NOT match CLIP-BPE — text encoded '
                'on this machine is incompatible with checkpoints trained '
                'with the standard merges table. Set DALLE_AMD_BPE_PATH or '
                'reinstall the package with its bundled data/ directory.',
                RuntimeWarning, stacklevel=2)

        symbols = list(self.byte_encoder.values())
        vocab = symbols + [s + '</w>' for s in symbols]
        vocab += [a + b for a, b in merges]
        # keep absolute id positions fixed even with a short/absent merge table
        vocab += [f'<unused{i}>' for i in range(self.N_MERGES - len(merges))]
        vocab += ['<|startoftext|>', '<|endoftext|>']

        self.vocab_size = self.VOCAB_SIZE
        self.encoder = {s: i for i, s in enumerate(vocab)}
        self.decoder = {i: s for s, i in self.encoder.items()}
        self.bpe_ranks = {m: r for r, m in enumerate(merges)}
        self._cache = {'<|startoftext|>': '<|startoftext|>',
                       '<|endoftext|>': '<|endoftext|>'}
        self.pattern = _re.compile(
            r"""<\|startoftext\|>|<\|endoftext\|>|'s|'t|'re|'ve|'m|'ll|'d|[\p{L}]+|[\p{N}]|[^\s\p{L}\p{N}]+"""
            if _re.__name__ == 'regex' else
            r"""<\|startoftext\|>|<\|endoftext\|>|'s|'t|'re|'ve|'m|'ll|'d|\w+|\d|[^\s\w\d]+""",
            _re.IGNORECASE)

    def bpe(self, token):
        if token in self._cache:
            return self._cache[token]
        word = tuple(token[:-1]) + (token[-1] + '</w>',)
        if len(word) == 1:
            return word[0]
        while len(word) > 1:
            pairs = _pairs(word)
            best = min(pairs, key=lambda p: self.bpe_ranks.get(p, float('inf')))
            if best not in self.bpe_ranks:
                break
            first, second = best
            merged = []
            i = 0
            while i < len(word):
                if i < len(word) - 1 and word[i] == first and word[i + 1] == second:
                    merged.append(first + second)
                    i += 2
                else:
                    merged.append(word[i])
                    i += 1
            word = tuple(merged)
        result = ' '.join(word)
        self._cache[token] = result
        return result

    def encode(self, text):
        ids = []
        text = _clean(text).lower()
        for token in self.pattern.findall(text):
            token = ''.join(self.byte_encoder[b] for b in token.encode('utf-8'))
            ids.extend(self.encoder[piece] for piece in self.bpe(token).split(' '))
        return ids

    def decode(self, tokens, remove_start_end=True, pad_tokens=set()):
        if torch.is_tensor(tokens):
            tokens = tokens.tolist()
        if remove_start_end:
            specials = {self.encoder['<|startoftext|>'], self.encoder['<|endoftext|>'], 0}
            tokens = [t for t in tokens if t not in specials]
        text = ''.join(self.decoder[t] for t in tokens if t not in pad_tokens)
        raw = bytearray(self.byte_decoder[c] for c in text
                        if c in self.byte_decoder)
        return raw.decode('utf-8', errors='replace').replace('</w>', ' ')

    def tokenize(self, texts, context_length=256, truncate_text=False):
        if isinstance(texts, str):
            texts = [texts]
        return _pad_batch([self.encode(t) for t in texts],
                          context_length, truncate_text, texts)


tokenizer = SimpleTokenizer()


class HugTokenizer:
    """HuggingFace tokenizers-json wrapper (reference tokenizer.py:158-192)."""

    def __init__(self, bpe_path=None):
        from tokenizers import Tokenizer
        from tokenizers.processors import ByteLevel
        path = Path(bpe_path)
        assert path.exists(), f'BPE json path {path} does not exist'
        tok = Tokenizer.from_file(str(path))
        tok.post_processor = ByteLevel(trim_offsets=True)
        self.tokenizer = tok
        self.vocab_size = tok.get_vocab_size()

    def decode(self, tokens, pad_tokens=set()):
        if torch.is_tensor(tokens):
            tokens = tokens.tolist()
        ignore = pad_tokens.union({0})
        return self.tokenizer.decode([t for t in tokens if t not in ignore],
                                     skip_special_tokens=True)

    def encode(self, text):
        return self.tokenizer.encode(text).ids

    def tokenize(self, texts, context_length=256, truncate_text=False):
        if isinstance(texts, str):
            texts = [texts]
        return _pad_batch([self.encode(t) for t in texts],
                          context_length, truncate_text, texts)


class ChineseTokenizer:
    """bert-base-chinese wordpiece (reference tokenizer.py:196-228). Accepts
    a local model directory since this environment has no network."""

    def __init__(self, model_name_or_path='bert-base-chinese'):
        from transformers import BertTokenizer
        self.tokenizer = BertTokenizer.from_pretrained(model_name_or_path)
        self.vocab_size = self.tokenizer.vocab_size

    def decode(self, tokens, pad_tokens=set()):
        if torch.is_tensor(tokens):
            tokens = tokens.tolist()
        ignore = pad_tokens.union({0})
        return self.tokenizer.decode([t for t in tokens if t not in ignore])

    def encode(self, text):
        return torch.tensor(self.tokenizer.encode(text, add_special_tokens=False))

    def tokenize(self, texts, context_length=256, truncate_text=False):
        if isinstance(texts, str):
            texts = [texts]
        return _pad_batch([self.encode(t).tolist() for t in texts],
                          context_length, truncate_text, texts)


class YttmTokenizer:
    """YouTokenToMe BPE-model wrapper (reference tokenizer.py:232-266)."""

    def __init__(self, bpe_path=None):
        try:
            import youtokentome as yttm
        except ImportError as e:
            raise ImportError(
                'youtokentome is not installed in this environment; use '
                'SimpleTokenizer or HugTokenizer instead') from e
        path = Path(bpe_path)
        assert path.exists(), f'BPE model path {path} does not exist'
        self._yttm = yttm
        self.tokenizer = yttm.BPE(model=str(path))
        self.vocab_size = self.tokenizer.vocab_size()

    def decode(self, tokens, pad_tokens=set()):
        if torch.is_tensor(tokens):
            tokens = tokens.tolist()
        return self.tokenizer.decode(tokens, ignore_ids=pad_tokens.union({0}))

    def encode(self, texts):
        ids = self.tokenizer.encode(texts, output_type=self._yttm.OutputType.ID)
        return list(map(torch.tensor, ids))

    def tokenize(self, texts, context_length=256, truncate_text=False):
        if isinstance(texts, str):
            texts = [texts]
        return _pad_batch([t.tolist() for t in self.encode(texts)],
                          context_length, truncate_text, texts)
