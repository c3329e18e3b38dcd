"""Paged KV cache for the decode engine (page = 64 positions);
backs the K4 agent-LLM decode of SURVEY.md 2.4.

Layouts match the HIP decode kernel (ops/hip/paged_attn.hip):
  K: [pages, KVH, D/8, 64, 8]  (d-major x8 -> coalesced MFMA A-frag reads)
  V: [pages, KVH, D, 64]       (transposed, pos minor -> coalesced MFMA
                                B-frag reads in the PV pass)

One allocator spans all layers: page p of layer l lives in that layer's
tensors, sharing the page-id space so a sequence has ONE page list used by
every layer (the usual vLLM-style design, sized for 288 GB HBM3E).
"""

from __future__ import annotations

import torch


class PagedKVCache:
    PAGE = 64

    def __init__(self, n_layers: int, kvh: int, d_head: int, n_pages: int,
                 device: str = "cuda", dtype=torch.bfloat16):
        self.n_layers = n_layers
        self.kvh = kvh
        self.d_head = d_head
        self.n_pages = n_pages
        self.device = device
        self.k = [torch.zeros(n_pages, kvh, d_head // 8, self.PAGE, 8,
                              dtype=dtype, device=device)
                  for _ in range(n_layers)]
        self.v = [torch.zeros(n_pages, kvh, d_head, self.PAGE, dtype=dtype,
                              device=device)
                  for _ in range(n_layers)]
        self._free = list(range(n_pages - 1, -1, -1))
        self._seq_pages: dict[int, list[int]] = {}
        self._seq_len: dict[int, int] = {}

    # ---- allocator -------------------------------------------------------
    @property
    def free_pages(self) -> int:
        return len(self._free)

    def pages_for(self, n_tokens: int) -> int:
        return (n_tokens + self.PAGE - 1) // self.PAGE

    def allocate(self, seq_id: int, n_tokens: int) -> None:
        need = self.pages_for(n_tokens)
        if need > len(self._free):
            raise MemoryError("KV cache out of pages")
        self._seq_pages[seq_id] = [self._free.pop() for _ in range(need)]
        self._seq_len[seq_id] = n_tokens

    def extend(self, seq_id: int, new_len: int) -> None:
        pages = self._seq_pages[seq_id]
        need = self.pages_for(new_len)
        while len(pages) < need:
            if not self._free:
                raise MemoryError("KV cache out of pages")
            pages.append(self._free.pop())
        self._seq_len[seq_id] = new_len

    def truncate(self, seq_id: int, new_len: int) -> None:
        """Roll a sequence back to new_len positions (prefix reuse across
        agent turns: decode-token KV beyond the shared prompt is discarded)."""
        pages = self._seq_pages[seq_id]
        keep = max(self.pages_for(new_len), 1)
        while len(pages) > keep:
            self._free.append(pages.pop())
        self._seq_len[seq_id] = new_len

    def free(self, seq_id: int) -> None:
        pages = self._seq_pages.pop(seq_id, [])
        self._free.extend(reversed(pages))
        self._seq_len.pop(seq_id, None)

    def seq_len(self, seq_id: int) -> int:
        return self._seq_len.get(seq_id, 0)

    # ---- tensors for a step ---------------------------------------------
    def block_table(self, seq_ids: list[int]) -> torch.Tensor:
        max_pages = max(len(self._seq_pages[s]) for s in seq_ids)
        bt = torch.zeros(len(seq_ids), max_pages, dtype=torch.int32)
        for i, s in enumerate(seq_ids):
            pages = self._seq_pages[s]
            bt[i, :len(pages)] = torch.tensor(pages, dtype=torch.int32)
        return bt.to(self.device)

    def seq_lens_tensor(self, seq_ids: list[int]) -> torch.Tensor:
        return torch.tensor([self._seq_len[s] for s in seq_ids],
                            dtype=torch.int32, device=self.device)

    def slot_ids(self, seq_id: int, start_pos: int, n: int) -> torch.Tensor:
        """Global slot id (page*64 + offset) for positions start..start+n."""
        pages = self._seq_pages[seq_id]
        slots = [pages[(start_pos + i) // self.PAGE] * self.PAGE +
                 (start_pos + i) % self.PAGE for i in range(n)]
        return torch.tensor(slots, dtype=torch.int32, device=self.device)
