"""SQLite-backed corpus (ref: src/data/corpus_sqlite.{h,cpp}).

For corpora too large for --shuffle-in-ram: sentences are loaded into an
SQLite table once, then every epoch streams them in a (seeded) random
order computed from SQLite's ORDER BY on a precomputed shuffle key, so
shuffling costs no host RAM.
"""
from __future__ import annotations

import os
import random
import sqlite3
from typing import Iterator, Sequence

from .corpus import Corpus, SentenceTuple
from .vocab import Vocab


class CorpusSQLite(Corpus):
    def __init__(self, paths: Sequence[str], vocabs: Sequence[Vocab],
                 db_path: str = "", drop: bool = False, **kw):
        super().__init__(paths, vocabs, **kw)
        self.db_path = db_path or (paths[0] + ".sqlite")
        if drop and os.path.exists(self.db_path):
            os.remove(self.db_path)
        self._ensure_db()

    def _connect(self):
        # a connection per use site: the BatchGenerator iterates from its
        # prefetch thread, and sqlite connections are thread-bound
        return sqlite3.connect(self.db_path)

    def _ensure_db(self):
        fresh = not os.path.exists(self.db_path)
        self.db = self._connect()
        n = len(self.paths) if not self.tsv else self.tsv_fields
        cols = ", ".join(f"s{i} TEXT" for i in range(n))
        self.db.execute(
            f"CREATE TABLE IF NOT EXISTS sents (pos INTEGER PRIMARY KEY, "
            f"{cols}, shuf REAL)")
        if fresh:
            ph = ", ".join("?" * (n + 2))
            rows = ((i, *r, random.random())
                    for i, r in enumerate(self._raw_lines()))
            self.db.executemany(f"INSERT INTO sents VALUES ({ph})", rows)
            self.db.execute("CREATE INDEX IF NOT EXISTS shuf_idx "
                            "ON sents(shuf)")
            self.db.commit()

    def iter_epoch(self, shuffle: bool = True,
                   skip: int = 0) -> Iterator[SentenceTuple]:
        db = self._connect()
        if shuffle:
            # re-key a seeded shuffle for this epoch (no host RAM)
            rng = random.Random(self.seed + self.epoch)
            db.execute("UPDATE sents SET shuf = "
                       "((pos * 2654435761 + ?) % 4294967296)",
                       (rng.randrange(1 << 32),))
            db.commit()
            cur = db.execute("SELECT * FROM sents ORDER BY shuf")
        else:
            cur = db.execute("SELECT * FROM sents ORDER BY pos")
        n_emitted = 0
        for row in cur:
            pos = row[0]
            texts = row[1:-1]
            tup = []
            ok = True
            for vi, text in enumerate(texts):
                vocab = self.vocabs[min(vi, len(self.vocabs) - 1)]
                ids = vocab.encode(text or "", add_eos=True)
                if len(ids) > self.max_length + 1:
                    if self.max_length_crop:
                        ids = ids[: self.max_length] + [vocab.eos_id]
                    else:
                        ok = False
                        break
                tup.append(ids)
            if not ok or any(len(t) == 0 for t in tup):
                continue
            n_emitted += 1
            if n_emitted <= skip:
                continue
            yield (pos, tup, {})
        self.epoch += 1
