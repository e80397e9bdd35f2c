"""LaTeX token vocabulary.

File-format contract preserved from the reference family (SURVEY.md §A.5;
reference `data_iterator.py::load_dict`, citation unverifiable — empty mount):
``dictionary.txt`` lines are ``"<token><ws><index>"``; index 0 is ``<eos>``.
"""

from __future__ import annotations

import os
from typing import Dict, List

EOS_ID = 0
EOS_TOKEN = "<eos>"


def load_dict(path: str) -> Dict[str, int]:
    """Read ``dictionary.txt`` → {token: index}.  Same contract as the
    reference's ``load_dict`` (whitespace-separated token/index lines)."""
    table: Dict[str, int] = {}
    with open(path, "r", encoding="utf-8") as f:
        for line in f:
            parts = line.split()
            if not parts:
                continue
            if len(parts) != 2:
                raise ValueError(f"bad dictionary line: {line!r}")
            tok, idx = parts[0], int(parts[1])
            if tok in table:
                raise ValueError(f"duplicate token {tok!r}")
            table[tok] = idx
    if table.get(EOS_TOKEN, None) != EOS_ID:
        # The family convention is <eos>=0; enforce it so decode semantics
        # (hypotheses die on token 0) are always valid.
        raise ValueError("dictionary must map '<eos>' to index 0")
    return table


class Vocab:
    """Bidirectional token <-> id map."""

    def __init__(self, token_to_id: Dict[str, int]):
        self.token_to_id = dict(token_to_id)
        n = max(self.token_to_id.values()) + 1
        self.id_to_token: List[str] = [""] * n
        for t, i in self.token_to_id.items():
            self.id_to_token[i] = t

    @classmethod
    def from_file(cls, path: str) -> "Vocab":
        return cls(load_dict(path))

    @classmethod
    def default(cls) -> "Vocab":
        """The repo's bundled 112-token CROHME-style dictionary."""
        here = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        return cls.from_file(os.path.join(here, "dictionary.txt"))

    def __len__(self) -> int:
        return len(self.id_to_token)

    def encode(self, caption: str) -> List[int]:
        """Whitespace-tokenized LaTeX → ids.  Unknown tokens raise (the
        reference family drops/errors on OOV; we error loudly)."""
        return [self.token_to_id[t] for t in caption.split()]

    def decode(self, ids) -> str:
        toks = []
        for i in ids:
            i = int(i)
            if i == EOS_ID:
                break
            toks.append(self.id_to_token[i])
        return " ".join(toks)
