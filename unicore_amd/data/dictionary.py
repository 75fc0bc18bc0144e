"""Symbol<->index vocabulary (parity: reference unicore/data/dictionary.py:12-148).

Text format: one ``<symbol> <count>`` pair per line, optional trailing
``#overwrite`` flag; special tokens default to the BERT-style
[CLS]/[PAD]/[SEP]/[UNK] set.
"""

import logging

import torch

logger = logging.getLogger(__name__)


class Dictionary:
    """A mapping from symbols to consecutive integers."""

    def __init__(
        self,
        *,  # keyword-only: the four roles are easy to transpose by accident
        bos="[CLS]",
        pad="[PAD]",
        eos="[SEP]",
        unk="[UNK]",
        extra_special_symbols=None,
    ):
        self.bos_word = bos
        self.pad_word = pad
        self.eos_word = eos
        self.unk_word = unk
        self.symbols = []   # index -> symbol
        self.count = []     # index -> corpus count
        self.indices = {}   # symbol -> index
        self.specials = {bos, pad, eos, unk}

    def __eq__(self, other):
        return other.indices == self.indices

    def __getitem__(self, idx):
        return self.symbols[idx] if idx < len(self.symbols) else self.unk_word

    def __len__(self) -> int:
        return len(self.indices)

    def __contains__(self, sym) -> bool:
        return sym in self.indices.keys()

    def index(self, sym) -> int:
        """Index of *sym*, falling back to the unk index."""
        assert isinstance(sym, str), "dictionary symbols are strings"
        found = self.indices.get(sym)
        return found if found is not None else self.unk()

    def vec_index(self, symbols) -> torch.Tensor:
        """Vectorize an iterable of symbols into a LongTensor of indices."""
        return torch.tensor([self.index(s) for s in symbols], dtype=torch.long)

    def special_index(self) -> list:
        """Indices of all registered special tokens."""
        return [self.index(s) for s in self.specials]

    def add_symbol(self, word, n=1, overwrite=False, is_special=False) -> int:
        """Register *word* (or bump its count when already known)."""
        if is_special:
            self.specials.add(word)
        if not overwrite and word in self.indices:
            known = self.indices[word]
            self.count[known] += n
            return known
        slot = len(self.symbols)
        self.indices[word] = slot
        self.symbols.append(word)
        self.count.append(n)
        return slot

    def pad_to_multiple_(self, padding_factor) -> None:
        """Grow the vocab with filler symbols until its size divides
        *padding_factor*. GEMM-shaped consumers (embedding matmuls, the
        lm-head projection, the fused cross entropy) are markedly faster
        when the vocab dimension is a multiple of 64."""
        if padding_factor <= 1:
            return
        filler = 0
        while len(self) % padding_factor:
            self.add_symbol(f"madeupword{filler:04d}", n=0, is_special=True)
            filler += 1

    # role helpers ---------------------------------------------------------

    def bos(self) -> int:
        """Index of the beginning-of-sentence symbol."""
        return self.index(sym=self.bos_word)

    def pad(self) -> int:
        """Index of the padding symbol."""
        return self.index(sym=self.pad_word)

    def eos(self) -> int:
        """Index of the end-of-sentence symbol."""
        return self.index(sym=self.eos_word)

    def unk(self) -> int:
        """Index of the unknown symbol."""
        assert self.unk_word in self.indices, "Cannot find unk symbol"
        return self.indices[self.unk_word]

    # file IO --------------------------------------------------------------

    @classmethod
    def load(cls, f):
        """Build a Dictionary from a ``<symbol> <count>``-per-line file."""
        built = cls()
        built.add_from_file(f)
        return built

    def add_from_file(self, f) -> None:
        """Merge symbols from an open file (or a path)."""
        if isinstance(f, str):
            try:
                with open(f, "r", encoding="utf-8") as handle:
                    self.add_from_file(handle)
            except UnicodeError:
                raise Exception(
                    f"Incorrect encoding detected in {f}, please rebuild"
                    " the dataset"
                )
            return

        raw_lines = f.readlines()
        for lineno, raw in enumerate(raw_lines):
            word, count, overwrite = self._parse_line(raw, raw_lines, lineno)
            if not overwrite and word in self:
                logger.info(
                    f"Duplicate word found when loading Dictionary: '{word}',"
                    f" index is {self.indices[word]}."
                )
            else:
                self.add_symbol(word, count, overwrite=overwrite)

    @staticmethod
    def _parse_line(raw, lines, lineno):
        try:
            head, _, tail = raw.rstrip().rpartition(" ")
            if not head:
                # bare symbol: synthesize a descending count so earlier
                # lines sort first
                head, tail = tail, str(len(lines) - lineno)
            overwrite = tail == "#overwrite"
            if overwrite:
                head, _, tail = head.rpartition(" ")
            return head, int(tail), overwrite
        except ValueError:
            raise ValueError(
                "Incorrect dictionary format, expected '<token> <cnt> [flags]'"
            )
