#!/usr/bin/env python3
"""Localize the HIP-enqueue E2E divergence (round-2 GPU call 1 found
test_forward_enqueue_matches_torch_path failing while every unit parity
passed).

Runs the failing scenario with instrumented dispatch: records the
(inputs, outputs) of every ops.enqueue_candidates call and every
MemoryBank.push call in both env modes, then diffs stage by stage:
  * do the two runs feed identical (feat, top1, gt)?   (else: upstream)
  * do enqueue_candidates outputs match?               (else: K5 rows kernel)
  * do bank states match after each push?              (else: bank_push)
Prints exact differing coordinates.
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import mgproto_amd.ops as O                     # noqa: E402
from mgproto_amd.model import construct_MGProto  # noqa: E402
from mgproto_amd.utils.memory import MemoryBank  # noqa: E402

records = []
_orig_enqueue = O.enqueue_candidates
_orig_push = MemoryBank.push


def rec_enqueue(feat, top1, gt, C, K, HW):
    f, l = _orig_enqueue(feat, top1, gt, C, K, HW)
    records.append(('enq',
                    feat.detach().cpu().clone(),
                    top1.detach().cpu().clone(), gt.detach().cpu().clone(),
                    f.detach().cpu().clone(), l.detach().cpu().clone(),
                    top1.dtype, top1.is_contiguous()))
    return f, l


def rec_push(self, feature, label):
    _orig_push(self, feature, label)
    records.append(('push', feature.detach().cpu().clone(),
                    label.detach().cpu().clone(),
                    self.mem.detach().cpu().clone(),
                    self.mem_len.detach().cpu().clone(),
                    self.head.detach().cpu().clone()))


def run(enable):
    global records
    records = []
    os.environ['MGPROTO_HIP_ENQUEUE'] = '1' if enable else '0'
    torch.manual_seed(0)
    C, K, d = 5, 3, 16
    m = construct_MGProto('resnet18', pretrained=False, img_size=64,
                          prototype_shape=(C * K, d, 1, 1),
                          num_classes=C, add_on_layers_type='regular',
                          sz_embedding=8, mem_capacity=8, mine_K=2).cuda()
    g = torch.Generator().manual_seed(3)
    x = torch.randn(6, 3, 64, 64, generator=g).cuda()
    y = (torch.arange(6) % C).cuda()
    with torch.no_grad():
        m(x, y)
        m(x.flip(0), y.flip(0))
    torch.cuda.synchronize()
    return records, (m.queue.mem.cpu(), m.queue.mem_len.cpu(),
                     m.queue.head.cpu())


def main():
    # monkeypatch both the model-level dispatch and the bank push
    import mgproto_amd.model as MM
    MM.ops.enqueue_candidates = rec_enqueue
    MemoryBank.push = rec_push

    rec_h, (mem_h, len_h, head_h) = run(True)
    rec_r, (mem_r, len_r, head_r) = run(False)

    print(f'records: hip={len(rec_h)} ref={len(rec_r)}')
    for i, (a, b) in enumerate(zip(rec_h, rec_r)):
        kind = a[0]
        assert kind == b[0], (i, a[0], b[0])
        if kind == 'enq':
            _, fa, ta, ga, oa, la, dt, cont = a
            _, fb, tb, gb, ob, lb, _, _ = b
            same_in = (torch.equal(fa, fb) and torch.equal(ta, tb)
                       and torch.equal(ga, gb))
            same_feat = torch.equal(oa, ob)
            same_lab = torch.equal(la, lb)
            print(f'[{i}] enq: inputs_equal={same_in} top1.dtype={dt} '
                  f'contig={cont} out_feat_equal={same_feat} '
                  f'out_lab_equal={same_lab}')
            if same_in and not (same_feat and same_lab):
                print('    hip lab:', la.tolist())
                print('    ref lab:', lb.tolist())
                if not same_feat:
                    df = (oa - ob).abs().amax(dim=1)
                    bad = torch.nonzero(df > 0).flatten().tolist()
                    print('    differing rows:', bad)
                    for r in bad[:6]:
                        print(f'      row {r}: hip={oa[r][:4].tolist()} '
                              f'ref={ob[r][:4].tolist()}')
            if not same_in:
                print('    INPUT DIVERGENCE (upstream of enqueue):',
                      'feat' if not torch.equal(fa, fb) else '',
                      'top1' if not torch.equal(ta, tb) else '',
                      'gt' if not torch.equal(ga, gb) else '')
        else:
            _, fa, la, ma, mla, ha = a
            _, fb, lb, mb, mlb, hb = b
            same_in = torch.equal(fa, fb) and torch.equal(la, lb)
            same_mem = torch.equal(ma, mb)
            print(f'[{i}] push: inputs_equal={same_in} '
                  f'mem_equal={same_mem} '
                  f'len_equal={torch.equal(mla, mlb)} '
                  f'head_equal={torch.equal(ha, hb)}')
            if same_in and not same_mem:
                d = (ma - mb).abs().amax(dim=2)      # [C, cap]
                for c, s in torch.nonzero(d > 0).tolist():
                    print(f'    mem[{c},{s}]: hip={ma[c, s][:4].tolist()} '
                          f'ref={mb[c, s][:4].tolist()}')
                # permutation check per class
                for c in range(ma.shape[0]):
                    sa = {tuple(r.tolist()) for r in ma[c]}
                    sb = {tuple(r.tolist()) for r in mb[c]}
                    if sa != sb:
                        print(f'    class {c}: row MULTISET differs '
                              f'(not a permutation)')
                    elif not torch.equal(ma[c], mb[c]):
                        print(f'    class {c}: same rows, different slots '
                              '(permutation)')

    print('final mem equal:', torch.equal(mem_h, mem_r),
          'len equal:', torch.equal(len_h, len_r),
          'head equal:', torch.equal(head_h, head_r))


if __name__ == '__main__':
    main()
