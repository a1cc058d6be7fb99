"""Inference/serving engine (no counterpart in the reference repo).

``InferenceEngine`` wraps an eval-mode MGProto for deployment:

* batched classification (softmax over the level-0 mixture logits),
* the OoD score p(x) = sum_c p(x, c) (reference train_and_test.py:199),
* prototype-level explanations: for the predicted class, the top
  activating prototypes with their mixture prior, peak activation, latent
  location and receptive-field bounding box in input pixels,
* optional hipGraph capture of the forward for fixed-batch low-latency
  serving (same machinery as the training-step capture).

``create_app`` exposes it over FastAPI (POST /predict with an image file;
GET /healthz, /model_info).
"""

import io
from typing import List, Optional

import torch
import torch.nn.functional as F

from .utils.receptive_field import compute_rf_prototype


class InferenceEngine:
    def __init__(self, model, device=None, amp: bool = True):
        self.model = model.eval()
        self.device = device or next(model.parameters()).device
        self.amp = amp and self.device.type == 'cuda'
        self._graph = None
        self._static_in = None
        self._static_out = None

    # ------------------------------------------------------------ forward
    def _forward(self, x: torch.Tensor):
        """(logits [B, C], distances [B, P, h, w]) for explanation."""
        ctx = (torch.autocast('cuda', dtype=torch.bfloat16)
               if self.amp else torch.no_grad())
        with torch.no_grad(), ctx:
            _, dist = self.model.push_forward(x)        # [B, P, h, w]
            acts = -dist
            B, P, H, W = acts.shape
            C = self.model.num_classes
            K = self.model.num_prototypes_per_class
            pooled = acts.view(B, P, H * W).max(dim=2).values   # [B, P]
            w = self.model.last_layer.weight                     # [C, P]
            diag = torch.arange(C, device=w.device)
            pi = w.view(C, C, K)[diag, diag]                     # [C, K]
            mixture = torch.einsum('bck,ck->bc',
                                   pooled.view(B, C, K), pi)     # [B, C]
            logits = torch.log(mixture.clamp_min(1e-30))
        return logits.float(), acts.float()

    def capture(self, batch_size: int, img_size: Optional[int] = None):
        """hipGraph-capture the fixed-batch forward (GPU serving)."""
        assert self.device.type == 'cuda'
        size = img_size or self.model.img_size
        self._static_in = torch.zeros(batch_size, 3, size, size,
                                      device=self.device).contiguous(
            memory_format=torch.channels_last)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self._forward(self._static_in)
        torch.cuda.current_stream().wait_stream(s)
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._static_out = self._forward(self._static_in)
        return self

    def _run(self, x: torch.Tensor):
        if self._graph is not None and x.shape == self._static_in.shape:
            self._static_in.copy_(x)
            self._graph.replay()
            return (self._static_out[0].clone(), self._static_out[1].clone())
        return self._forward(x)

    # ------------------------------------------------------------ predict
    @torch.no_grad()
    def predict(self, images: torch.Tensor, topk_classes: int = 5,
                explain_topk: int = 3) -> List[dict]:
        """images: [B, 3, H, W] normalized tensor -> one dict per image."""
        x = images.to(self.device)
        if self.device.type == 'cuda':
            x = x.contiguous(memory_format=torch.channels_last)
        logits, acts = self._run(x)
        probs = F.softmax(logits, dim=1)
        density = torch.exp(logits).sum(dim=1)          # p(x), OoD score
        B, P, H, W = acts.shape
        K = self.model.num_prototypes_per_class
        k_cls = min(topk_classes, probs.shape[1])
        top_p, top_c = probs.topk(k_cls, dim=1)

        results = []
        for b in range(B):
            pred = int(top_c[b, 0])
            expl = self._explain(acts[b], pred, K, H, W, explain_topk)
            results.append({
                'pred_class': pred,
                'top_classes': [{'class': int(c), 'prob': float(p)}
                                for c, p in zip(top_c[b], top_p[b])],
                'density_pX': float(density[b]),
                'explanations': expl,
            })
        return results

    def _explain(self, acts_b: torch.Tensor, cls: int, K: int, H: int,
                 W: int, topk: int) -> List[dict]:
        own = acts_b[cls * K:(cls + 1) * K]              # [K, H, W]
        peak, arg = own.view(K, H * W).max(dim=1)
        order = peak.argsort(descending=True)[:min(topk, K)]
        w = self.model.last_layer.weight
        out = []
        for k in order.tolist():
            h, wdt = int(arg[k]) // W, int(arg[k]) % W
            entry = {'prototype': cls * K + k,
                     'prior': float(w[cls, cls * K + k]),
                     'activation': float(peak[k]),
                     'latent_hw': [h, wdt]}
            info = getattr(self.model, 'proto_layer_rf_info', None)
            if info is not None and h < info[0] and wdt < info[0]:
                box = compute_rf_prototype(self.model.img_size,
                                           [0, h, wdt], info)
                entry['rf_bbox_yxyx'] = box[1:]
            out.append(entry)
        return out


# ---------------------------------------------------------------------------
# FastAPI app
# ---------------------------------------------------------------------------

def create_app(engine: InferenceEngine, class_names: Optional[List[str]] = None):
    from fastapi import FastAPI, Request
    from mgproto_amd.data import transforms as T
    from mgproto_amd.data.preprocess import mean, std

    app = FastAPI(title='mgproto_amd', version='0.1')
    tf = T.Compose([T.Resize((engine.model.img_size, engine.model.img_size)),
                    T.ToTensor(), T.Normalize(mean, std)])

    @app.get('/healthz')
    def healthz():
        return {'status': 'ok', 'device': str(engine.device)}

    @app.get('/model_info')
    def model_info():
        m = engine.model
        return {'num_classes': m.num_classes,
                'num_prototypes': m.num_prototypes,
                'prototypes_per_class': m.num_prototypes_per_class,
                'img_size': m.img_size,
                'graph_captured': engine._graph is not None}

    @app.post('/predict')
    async def predict(request: Request, topk: int = 5, explain: int = 3):
        # raw image bytes as the request body (multipart would need the
        # python-multipart package, absent in this environment)
        from PIL import Image
        raw = await request.body()
        img = Image.open(io.BytesIO(raw)).convert('RGB')
        x = tf(img).unsqueeze(0)
        res = engine.predict(x, topk_classes=topk, explain_topk=explain)[0]
        if class_names is not None:
            for t in res['top_classes']:
                t['name'] = class_names[t['class']]
        return res

    return app
