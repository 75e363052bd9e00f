"""Tokenizers (reference training/tokenizer/tokenizer.py, ~900 LoC).

Builds Null / GPT2-BPE / SentencePiece / HuggingFace tokenizers.  The
MegaScope additions are kept: ``decoder`` and ``offsets`` properties used
by the visualization server (reference :157,166,378-385,479-527).
"""

from __future__ import annotations

from typing import List, Optional


class MegatronTokenizer:
    def tokenize(self, text: str) -> List[int]:
        raise NotImplementedError

    def detokenize(self, ids: List[int]) -> str:
        raise NotImplementedError

    @property
    def vocab_size(self) -> int:
        raise NotImplementedError

    @property
    def eod(self) -> int:
        raise NotImplementedError

    @property
    def decoder(self):
        """id -> printable token map (MegaScope)."""
        return {}

    def offsets(self, ids: List[int], text: str) -> List[int]:
        offsets, pos = [], 0
        for i in ids:
            offsets.append(pos)
            pos += len(self.detokenize([i]))
        return offsets


class NullTokenizer(MegatronTokenizer):
    """Integer-string tokenizer for synthetic data (reference NullTokenizer)."""

    def __init__(self, vocab_size: int = 131072):
        self._vocab_size = int(vocab_size)
        self._eod = self._vocab_size - 1

    def tokenize(self, text):
        return [int(t) for t in text.split()]

    def detokenize(self, ids):
        return " ".join(str(i) for i in ids)

    @property
    def vocab_size(self):
        return self._vocab_size

    @property
    def eod(self):
        return self._eod

    @property
    def decoder(self):
        return _LazyIntDecoder()


class _LazyIntDecoder(dict):
    def get(self, key, default=None):
        return str(key)

    def __getitem__(self, key):
        return str(key)

    def __contains__(self, key):
        return True


class HuggingFaceTokenizer(MegatronTokenizer):
    def __init__(self, model_name_or_path: str):
        from transformers import AutoTokenizer
        self._tok = AutoTokenizer.from_pretrained(model_name_or_path)

    def tokenize(self, text):
        return self._tok.encode(text)

    def detokenize(self, ids):
        return self._tok.decode(ids)

    @property
    def vocab_size(self):
        return len(self._tok)

    @property
    def eod(self):
        return self._tok.eos_token_id

    @property
    def decoder(self):
        return _LazyReadableDecoder(self._tok)


class _LazyReadableDecoder(dict):
    """Decodes ids on demand into readable tokens (reference :321)."""

    def __init__(self, tok):
        super().__init__()
        self._tok = tok

    def get(self, key, default=None):
        try:
            return self._tok.decode([int(key)])
        except Exception:  # noqa: BLE001
            return default

    def __getitem__(self, key):
        return self._tok.decode([int(key)])


class GPT2BPETokenizer(MegatronTokenizer):
    def __init__(self, vocab_file: str, merge_file: str):
        from tokenizers import ByteLevelBPETokenizer
        self._tok = ByteLevelBPETokenizer(vocab_file, merge_file)
        self._eod = self._tok.token_to_id("<|endoftext|>")
        if self._eod is None:
            self._eod = self._tok.get_vocab_size() - 1

    def tokenize(self, text):
        return self._tok.encode(text).ids

    def detokenize(self, ids):
        return self._tok.decode(ids)

    @property
    def vocab_size(self):
        return self._tok.get_vocab_size()

    @property
    def eod(self):
        return self._eod

    @property
    def decoder(self):
        return {i: self._tok.id_to_token(i) or str(i)
                for i in range(self.vocab_size)}


class SentencePieceTokenizer(MegatronTokenizer):
    def __init__(self, model_file: str):
        import sentencepiece as spm
        self._tok = spm.SentencePieceProcessor(model_file=model_file)

    def tokenize(self, text):
        return self._tok.encode(text)

    def detokenize(self, ids):
        return self._tok.decode(ids)

    @property
    def vocab_size(self):
        return self._tok.get_piece_size()

    @property
    def eod(self):
        return self._tok.eos_id()


class BertWordPieceTokenizer(MegatronTokenizer):
    """WordPiece over a BERT vocab.txt (reference tokenizer.py
    BertWordPieceLowerCase / BertWordPieceCase)."""

    def __init__(self, vocab_file: str, lower_case: bool = True):
        from tokenizers import BertWordPieceTokenizer as _BWP
        self._tok = _BWP(vocab_file, lowercase=lower_case)
        v = self._tok.get_vocab()
        self._cls = v.get("[CLS]", 0)
        self._sep = v.get("[SEP]", 0)
        self._pad = v.get("[PAD]", 0)
        self._mask = v.get("[MASK]", 0)

    def tokenize(self, text):
        return self._tok.encode(text, add_special_tokens=False).ids

    def detokenize(self, ids):
        return self._tok.decode(ids)

    @property
    def vocab_size(self):
        return self._tok.get_vocab_size()

    @property
    def vocab(self):
        return self._tok.get_vocab()

    @property
    def cls(self):
        return self._cls

    @property
    def sep(self):
        return self._sep

    @property
    def pad(self):
        return self._pad

    @property
    def mask(self):
        return self._mask

    @property
    def eod(self):
        return self._sep


class TikTokenizer(MegatronTokenizer):
    """Native byte-level BPE over a .tiktoken mergeable-ranks file
    ("<base64> <rank>" lines — the reference's TikTokenizer via the
    tiktoken package, reimplemented so no extra dependency is needed)."""

    PATTERN = (r"""'(?i:[sdmt]|ll|ve|re)|[^\r\n\p{L}\p{N}]?+\p{L}+"""
               r"""|\p{N}{1,3}| ?[^\s\p{L}\p{N}]++[\r\n]*"""
               r"""|\s*[\r\n]|\s+(?!\S)|\s+""")

    def __init__(self, model_file: str, special_tokens=None):
        import base64
        import regex
        self._ranks = {}
        with open(model_file, "rb") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                tok_b64, rank = line.split()
                self._ranks[base64.b64decode(tok_b64)] = int(rank)
        self._pat = regex.compile(self.PATTERN)
        n = len(self._ranks)
        self._specials = {}
        for i, name in enumerate(special_tokens or ["<|endoftext|>"]):
            self._specials[name] = n + i
        self._eod = self._specials.get("<|endoftext|>", n)
        self._id_to_bytes = {r: b for b, r in self._ranks.items()}
        for name, i in self._specials.items():
            self._id_to_bytes[i] = name.encode()

    def _bpe(self, piece: bytes):
        parts = [piece[i:i + 1] for i in range(len(piece))]
        while len(parts) > 1:
            best, best_rank = None, None
            for i in range(len(parts) - 1):
                r = self._ranks.get(parts[i] + parts[i + 1])
                if r is not None and (best_rank is None or r < best_rank):
                    best, best_rank = i, r
            if best is None:
                break
            parts[best:best + 2] = [parts[best] + parts[best + 1]]
        return [self._ranks[p] for p in parts if p in self._ranks]

    def tokenize(self, text):
        out = []
        for piece in self._pat.findall(text):
            b = piece.encode("utf-8")
            if b in self._ranks:
                out.append(self._ranks[b])
            else:
                out.extend(self._bpe(b))
        return out

    def detokenize(self, ids):
        data = b"".join(self._id_to_bytes.get(i, b"") for i in ids)
        return data.decode("utf-8", errors="replace")

    @property
    def vocab_size(self):
        return len(self._ranks) + len(self._specials)

    @property
    def eod(self):
        return self._eod


def build_tokenizer(args):
    t = args.tokenizer_type
    if t == "NullTokenizer":
        return NullTokenizer(args.padded_vocab_size or args.vocab_size or 131072)
    if t == "GPT2BPETokenizer":
        return GPT2BPETokenizer(args.vocab_file, args.merge_file)
    if t == "SentencePieceTokenizer":
        return SentencePieceTokenizer(args.tokenizer_model)
    if t == "HuggingFaceTokenizer":
        return HuggingFaceTokenizer(args.tokenizer_model)
    if t == "BertWordPieceLowerCase":
        return BertWordPieceTokenizer(args.vocab_file, lower_case=True)
    if t == "BertWordPieceCase":
        return BertWordPieceTokenizer(args.vocab_file, lower_case=False)
    if t == "TikTokenizer":
        return TikTokenizer(args.tokenizer_model)
    raise ValueError(f"unknown tokenizer type {t}")
