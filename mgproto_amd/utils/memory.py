"""Class-wise FIFO memory bank of patch features.

Re-design of the reference MemoryBank (``/root/reference/utils/memory.py``):

* storage is ONE device-resident ``[C, cap, d]`` buffer + per-class length
  and ring head, instead of 200 separate ``cls{i}`` buffers — one scatter
  writes a whole batch (the reference loops classes in Python,
  memory.py:48-71, and shifts the whole buffer on overflow, memory.py:64);
* ``push`` is fully batched and deterministic (stable sort by class, FIFO
  ring positions) with no host synchronization;
* the state-dict surface still presents the reference layout
  (``cls0..cls{C-1}`` of shape [cap, d] in oldest-first order plus
  ``mem_len``), so checkpoints interoperate both ways.

Semantics relative to the reference: the stored multiset per class is
identical (FIFO, capacity cap); rows may sit at different physical offsets
(ring vs shift) which is unobservable through ``pull_*``/state_dict (both
return logical order). One divergence: when a single push exceeds cap the
reference keeps a *random* subsample (memory.py:52) — we keep the newest
cap items (deterministic; unreachable with default shapes since
B*K <= cap).
"""

import os
from typing import Optional, Tuple

import torch
import torch.nn as nn


class MemoryBank(nn.Module):
    def __init__(self, num_classes: int, dim_feature: int, capacity: int = 1024,
                 mode: str = 'all', fix_length_mult: int = 4):
        super().__init__()
        assert capacity % num_classes == 0, (capacity, num_classes)
        self.num_classes = num_classes
        self.dim_feature = dim_feature
        self.capacity = capacity
        self.cap_cls = capacity // num_classes
        self.mode = mode
        self.fix_length_mult = fix_length_mult

        # flat storage with one extra "trash" class row: rows with the
        # sentinel label ``num_classes`` land there, which lets the
        # distributed enqueue push fixed-size padded batches with no host
        # sync (invalid rows are written to trash and never read)
        self.register_buffer('_mem_flat',
                             torch.zeros((num_classes + 1) * self.cap_cls,
                                         dim_feature))
        self.register_buffer('mem_len', torch.zeros(num_classes, dtype=torch.int64))
        self.register_buffer('head', torch.zeros(num_classes, dtype=torch.int64))

    @property
    def mem(self) -> torch.Tensor:
        """[C, cap, d] view of the live storage (excludes the trash row)."""
        return self._mem_flat[:self.num_classes * self.cap_cls].view(
            self.num_classes, self.cap_cls, self.dim_feature)

    # ------------------------------------------------------------------ push
    @torch.no_grad()
    def push(self, feature: torch.Tensor, label: torch.Tensor) -> None:
        """Append (feature, label) pairs FIFO per class. Batched, in-order.

        ``feature``: [M, d]; ``label``: [M] int64. Items of one class are
        inserted in their order of appearance. Rows labelled with the
        sentinel ``num_classes`` are discarded (written to the trash row) —
        no host sync, used by padded distributed pushes.
        """
        assert feature.dim() == 2 and label.dim() == 1
        assert feature.size(0) == label.size(0)
        M = feature.size(0)
        if M == 0:
            return
        feature = feature.detach()
        label = label.detach()
        cap = self.cap_cls
        C = self.num_classes
        dev = feature.device

        if (feature.is_cuda and feature.dtype == torch.float32
                and os.environ.get('MGPROTO_HIP_ENQUEUE') == '1'):
            # K5 HIP path (opt-in; round-2 parity-validated on MI355X,
            # throughput-neutral vs this torch path): deterministic
            # class-segregated ring write, same bank state as the torch
            # sort/scan/scatter below
            try:
                from ..ops import hip_loader
                ext = hip_loader.load()
            except Exception:  # noqa: BLE001
                ext = None
            if ext is not None:
                ext.bank_push(feature.contiguous(),
                              label.to(torch.int64).contiguous(),
                              self._mem_flat, self.head, self.mem_len,
                              C, cap)
                return

        order = torch.argsort(label, stable=True)
        lab = label[order]                    # sentinel rows sort last
        fea = feature[order]

        # position within each class segment (cummax of segment starts —
        # boolean compaction would be a host sync and break graph capture)
        change = torch.ones(M, dtype=torch.bool, device=dev)
        change[1:] = lab[1:] != lab[:-1]
        idx = torch.arange(M, device=dev)
        seg_start = torch.cummax(torch.where(change, idx,
                                             torch.zeros_like(idx)), 0)[0]
        within = idx - seg_start                                    # [M]

        counts = torch.zeros(C + 1, dtype=torch.int64, device=dev)
        counts.scatter_add_(0, lab, torch.ones_like(lab))

        # if a class pushes more than cap at once, write only the newest cap
        # (stale rows' flat indices are redirected to the trash row so the
        # whole batch scatters in one op; the advanced head still points at
        # the oldest surviving row)
        keep = within >= (counts[lab] - cap)
        is_trash = (lab == C) | ~keep
        pos = (self.head[torch.clamp(lab, max=C - 1)] + within) % cap
        flat = torch.where(is_trash,
                           C * cap + (within % cap),
                           torch.clamp(lab, max=C - 1) * cap + pos)
        self._mem_flat[flat] = fea.to(self._mem_flat.dtype)

        counts_real = counts[:C]
        self.head = (self.head + counts_real) % cap
        self.mem_len = torch.minimum(self.mem_len + counts_real,
                                     torch.full_like(self.mem_len, cap))

    # ------------------------------------------------------------------ pull
    def _logical(self, c: int) -> torch.Tensor:
        """Rows of class c, oldest first."""
        L = int(self.mem_len[c])
        h = int(self.head[c])
        if L < self.cap_cls:
            return self.mem[c, :L]
        return torch.cat([self.mem[c, h:], self.mem[c, :h]], dim=0)

    @torch.no_grad()
    def pull_all(self) -> Tuple[Optional[torch.Tensor], Optional[torch.Tensor]]:
        """All stored features + labels (reference memory.py:136-151)."""
        lens = self.mem_len
        if int(lens.sum()) == 0:
            return None, None
        out_data, out_label = [], []
        for c in range(self.num_classes):
            L = int(lens[c])
            if L == 0:
                continue
            out_data.append(self._logical(c))
            out_label.append(torch.full((L,), c, dtype=torch.int64,
                                        device=self.mem.device))
        return torch.cat(out_data, 0), torch.cat(out_label, 0)

    @torch.no_grad()
    def pull_dense(self, classes: torch.Tensor) -> torch.Tensor:
        """[G, cap, d] stack for the given (full) classes — the batched-EM
        feed. Requires every requested class to be full; physical order is
        used (EM is permutation-invariant over samples)."""
        return self.mem.index_select(0, classes)

    def full_mask(self) -> torch.Tensor:
        return self.mem_len == self.cap_cls

    @torch.no_grad()
    def pull_fix_class(self, label: torch.Tensor):
        """Reference memory.py:76-93: features of the flagged classes."""
        assert list(label.size()) == [self.num_classes]
        indices = torch.nonzero(label.detach(), as_tuple=False).flatten()
        out_data, out_label = [], []
        for i in indices.tolist():
            L = int(self.mem_len[i])
            if L == 0:
                continue
            out_data.append(self._logical(i))
            out_label.append(torch.full((L,), i, dtype=torch.int64,
                                        device=self.mem.device))
        if not out_data:
            return None, None
        return torch.cat(out_data, 0), torch.cat(out_label, 0)

    @torch.no_grad()
    def pull_fix_length(self, label: torch.Tensor):
        """Reference memory.py:95-133: per query row, a fixed-length sample
        of features whose classes are flagged in that row's one-hot."""
        assert label.dim() == 2 and label.size(1) == self.num_classes
        pull_num = self.cap_cls * self.fix_length_mult
        label = label.detach()
        indices = torch.nonzero(label.max(0)[0], as_tuple=False).flatten()
        cand_data, cand_label = [], []
        for i in indices.tolist():
            L = int(self.mem_len[i])
            if L == 0:
                continue
            cand_data.append(self._logical(i))
            cand_label.append(torch.full((L,), i, dtype=torch.int64,
                                         device=self.mem.device))
        if not cand_data:
            return None, None
        cand_data = torch.cat(cand_data, 0)
        cand_label = torch.cat(cand_label, 0)
        oh = torch.nn.functional.one_hot(cand_label, self.num_classes).float()
        cand_mask = torch.matmul(label.float(), oh.T) > 0.999
        if cand_mask.sum(1).min() < 1:
            return None, None
        out = []
        for i in range(label.size(0)):
            data = cand_data[cand_mask[i]]
            while data.size(0) < pull_num:
                data = data.repeat(2, 1)
            if data.size(0) > pull_num:
                data = data[torch.randperm(data.size(0))[:pull_num]]
            out.append(data.unsqueeze(1))
        return torch.cat(out, 1).contiguous(), None

    def pull(self, *args, **kwargs):
        if self.mode == 'all':
            return self.pull_all(*args, **kwargs)
        return getattr(self, 'pull_' + self.mode)(*args, **kwargs)

    # ------------------------------------------------- reference state_dict
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        # Export the reference layout: cls{i} [cap, d] oldest-first + mem_len.
        for c in range(self.num_classes):
            buf = torch.zeros_like(self.mem[c])
            L = int(self.mem_len[c])
            if L > 0:
                buf[:L] = self._logical(c)[:L] if L < self.cap_cls else self._logical(c)
            destination[prefix + 'cls%d' % c] = buf
        destination[prefix + 'mem_len'] = (self.mem_len if keep_vars
                                           else self.mem_len.detach().clone())

    def _load_from_state_dict(self, state_dict, prefix, local_metadata, strict,
                              missing_keys, unexpected_keys, error_msgs):
        found = 0
        for c in range(self.num_classes):
            key = prefix + 'cls%d' % c
            if key in state_dict:
                src = state_dict[key]
                if src.shape != self.mem[c].shape:
                    error_msgs.append(
                        f'size mismatch for {key}: checkpoint has '
                        f'{tuple(src.shape)}, bank capacity is '
                        f'{tuple(self.mem[c].shape)} — construct the model '
                        'with the checkpoint\'s mem_capacity (drivers infer '
                        'it via infer_ctor_kwargs_from_state)')
                    state_dict.pop(key)
                    continue
                self.mem[c].copy_(src)
                found += 1
                state_dict.pop(key)
            elif strict:
                missing_keys.append(key)
        key = prefix + 'mem_len'
        if key in state_dict:
            self.mem_len.copy_(state_dict[key])
            state_dict.pop(key)
        elif strict:
            missing_keys.append(key)
        # logical order was exported oldest-first, so heads reset
        self.head.copy_(self.mem_len % self.cap_cls)
        if strict and found not in (0, self.num_classes):
            error_msgs.append(f'MemoryBank: only {found}/{self.num_classes} '
                              'cls buffers present in checkpoint')
