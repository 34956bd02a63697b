"""Tokenizer loading for the serving path.

Wraps whichever artifact ships with the user's checkpoint directory:
HF `tokenizer.json` (tokenizers lib) or a sentencepiece `*.model`.
The reference launches user code and never tokenizes; this exists so the
in-framework serving example (`Llama.generate`) works end-to-end on real
checkpoints.
"""
import os


class Tokenizer:
    def __init__(self, backend, kind):
        self._t = backend
        self.kind = kind  # "hf" | "sp"

    @classmethod
    def load(cls, path):
        """path: a tokenizer file or a checkpoint dir containing one."""
        if os.path.isdir(path):
            for cand in ("tokenizer.json", "tokenizer.model"):
                p = os.path.join(path, cand)
                if os.path.exists(p):
                    path = p
                    break
            else:
                raise FileNotFoundError(f"no tokenizer artifact in {path}")
        if path.endswith(".json"):
            from tokenizers import Tokenizer as HFTok

            return cls(HFTok.from_file(path), "hf")
        import sentencepiece as spm

        sp = spm.SentencePieceProcessor()
        sp.Load(path)
        return cls(sp, "sp")

    def encode(self, text, add_bos=False):
        if self.kind == "hf":
            ids = self._t.encode(text).ids
        else:
            ids = self._t.EncodeAsIds(text)
        if add_bos and self.bos_id is not None:
            ids = [self.bos_id] + ids
        return ids

    def decode(self, ids):
        if self.kind == "hf":
            return self._t.decode(list(ids))
        return self._t.DecodeIds(list(ids))

    @property
    def vocab_size(self):
        if self.kind == "hf":
            return self._t.get_vocab_size()
        return self._t.GetPieceSize()

    @property
    def bos_id(self):
        if self.kind == "hf":
            tok = self._t.token_to_id("<s>") or self._t.token_to_id(
                "<|begin_of_text|>")
            return tok
        b = self._t.bos_id()
        return b if b >= 0 else None

    @property
    def eos_id(self):
        if self.kind == "hf":
            return (self._t.token_to_id("</s>")
                    or self._t.token_to_id("<|end_of_text|>"))
        e = self._t.eos_id()
        return e if e >= 0 else None
