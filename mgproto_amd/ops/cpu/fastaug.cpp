// Native CPU augmentation core: homography warp + color jitter + normalize
// in ONE pass over output pixels.
//
// The Python fused pipeline (data/transforms.py FusedTrainTransform) already
// collapses the reference's four geometric resamples into one homography;
// this C++ core additionally fuses the PIL resample, the color jitter and
// the ToTensor+Normalize passes, so each output pixel is touched once.
// Python draws the random parameters (identical distribution); this op is
// pure deterministic math. Used by the dataloader workers (SURVEY.md hard
// part #5: the reference's input path cannot feed one MI355X, let alone 8).

#include <torch/extension.h>
#include <cmath>
#include <cstdint>
#include <algorithm>

namespace {

inline float clampf(float v, float lo, float hi) {
    return v < lo ? lo : (v > hi ? hi : v);
}

// PIL-convention homography: output (x, y) -> source coords
struct H33 {
    double a, b, c, d, e, f, g, h;
    inline void map(double x, double y, double& u, double& v) const {
        const double w = g * x + h * y + 1.0;
        u = (a * x + b * y + c) / w;
        v = (d * x + e * y + f) / w;
    }
};

// bilinear sample with zero fill outside (PIL Image.transform semantics)
inline void sample(const uint8_t* img, int Hh, int Ww, double u, double v,
                   float out[3]) {
    const int x0 = (int)std::floor(u), y0 = (int)std::floor(v);
    const double fx = u - x0, fy = v - y0;
    float acc[3] = {0.f, 0.f, 0.f};
    for (int dy = 0; dy < 2; ++dy) {
        const int yy = y0 + dy;
        const double wy = dy ? fy : 1.0 - fy;
        if (yy < 0 || yy >= Hh || wy == 0.0) continue;
        for (int dx = 0; dx < 2; ++dx) {
            const int xx = x0 + dx;
            const double wx = dx ? fx : 1.0 - fx;
            if (xx < 0 || xx >= Ww || wx == 0.0) continue;
            const float w = (float)(wx * wy);
            const uint8_t* p = img + ((size_t)yy * Ww + xx) * 3;
            acc[0] += w * p[0];
            acc[1] += w * p[1];
            acc[2] += w * p[2];
        }
    }
    out[0] = acc[0]; out[1] = acc[1]; out[2] = acc[2];
}

inline void rgb_to_hsv(float r, float g, float b, float& h, float& s, float& v) {
    const float mx = std::max(r, std::max(g, b));
    const float mn = std::min(r, std::min(g, b));
    v = mx;
    const float d = mx - mn;
    s = mx == 0.f ? 0.f : d / mx;
    if (d == 0.f) { h = 0.f; return; }
    if (mx == r) h = (g - b) / d + (g < b ? 6.f : 0.f);
    else if (mx == g) h = (b - r) / d + 2.f;
    else h = (r - g) / d + 4.f;
    h /= 6.f;
}

inline void hsv_to_rgb(float h, float s, float v, float& r, float& g, float& b) {
    if (s <= 0.f) { r = g = b = v; return; }
    h = h - std::floor(h);
    const float hh = h * 6.f;
    const int i = (int)hh;
    const float f = hh - i;
    const float p = v * (1.f - s);
    const float q = v * (1.f - s * f);
    const float t = v * (1.f - s * (1.f - f));
    switch (i % 6) {
        case 0: r = v; g = t; b = p; break;
        case 1: r = q; g = v; b = p; break;
        case 2: r = p; g = v; b = t; break;
        case 3: r = p; g = q; b = v; break;
        case 4: r = t; g = p; b = v; break;
        default: r = v; g = p; b = q; break;
    }
}

}  // namespace

// img_u8: [H, W, 3] uint8; coeffs: [8] float64 (PIL homography);
// jitter: (brightness, contrast, saturation, hue_shift) factors — 1/1/1/0
// disables; order: permutation code 0..5 of (b, c, s) application order;
// mean/std: [3] normalization. Returns [3, S, S] float32 CHW.
torch::Tensor warp_jitter_normalize(torch::Tensor img_u8, torch::Tensor coeffs,
                                    int64_t out_size,
                                    double bright, double contrast,
                                    double satur, double hue_shift,
                                    int64_t order,
                                    torch::Tensor mean, torch::Tensor stdv) {
    TORCH_CHECK(img_u8.dtype() == torch::kUInt8 && img_u8.dim() == 3
                && img_u8.size(2) == 3, "img must be [H, W, 3] uint8");
    img_u8 = img_u8.contiguous();
    coeffs = coeffs.to(torch::kFloat64).contiguous();
    TORCH_CHECK(coeffs.numel() == 8, "coeffs must have 8 elements");
    TORCH_CHECK(out_size > 0, "out_size must be positive");
    TORCH_CHECK(mean.numel() == 3 && stdv.numel() == 3,
                "mean/std must have 3 elements");
    const int Hh = img_u8.size(0), Ww = img_u8.size(1);
    const int S = (int)out_size;
    const double* cf = coeffs.data_ptr<double>();
    H33 M{cf[0], cf[1], cf[2], cf[3], cf[4], cf[5], cf[6], cf[7]};
    const uint8_t* src = img_u8.data_ptr<uint8_t>();

    auto out = torch::empty({3, S, S}, torch::kFloat32);
    float* dst = out.data_ptr<float>();
    auto buf = torch::empty({S, S, 3}, torch::kFloat32);
    float* wb = buf.data_ptr<float>();

    // pass 1: warp (and luminance sum for the contrast op)
    double lum_sum = 0.0;
    at::parallel_for(0, S, 8, [&](int64_t y0, int64_t y1) {
        for (int64_t y = y0; y < y1; ++y) {
            for (int x = 0; x < S; ++x) {
                double u, v;
                // PIL convention: map at the output pixel CENTER, sample a
                // half-pixel back (Geometry.c) — keeps bit-level parity
                // with the Image.transform fallback path
                M.map((double)x + 0.5, (double)y + 0.5, u, v);
                sample(src, Hh, Ww, u - 0.5, v - 0.5,
                       wb + (y * S + x) * 3);
            }
        }
    });
    const bool do_c = contrast != 1.0;
    if (do_c) {
        for (int64_t i = 0; i < (int64_t)S * S; ++i)
            lum_sum += 0.299 * wb[i * 3] + 0.587 * wb[i * 3 + 1]
                + 0.114 * wb[i * 3 + 2];
    }
    const float lmean = (float)(lum_sum / ((double)S * S));

    const bool do_b = bright != 1.0;
    const bool do_s = satur != 1.0;
    const bool do_h = hue_shift != 0.0;
    const float bf = (float)bright, cfc = (float)contrast, sf = (float)satur;
    const float hs = (float)hue_shift;
    // order: 0=bcs 1=bsc 2=cbs 3=csb 4=sbc 5=scb (hue always last)
    const int seq[6][3] = {{0, 1, 2}, {0, 2, 1}, {1, 0, 2},
                           {1, 2, 0}, {2, 0, 1}, {2, 1, 0}};
    const int* op = seq[order % 6];
    // the contrast pivot is the luminance mean AT THE TIME contrast runs:
    // a preceding brightness op scales it by bf (saturation preserves
    // per-pixel luminance, so it never changes the mean)
    float lmean_c = lmean;
    if (do_b && do_c) {
        for (int k = 0; k < 3; ++k) {
            if (op[k] == 0) { lmean_c = lmean * bf; break; }
            if (op[k] == 1) break;
        }
    }
    const float nm[3] = {mean[0].item<float>(), mean[1].item<float>(),
                         mean[2].item<float>()};
    const float ns[3] = {stdv[0].item<float>(), stdv[1].item<float>(),
                         stdv[2].item<float>()};

    at::parallel_for(0, S, 8, [&](int64_t y0, int64_t y1) {
        for (int64_t y = y0; y < y1; ++y) {
            for (int x = 0; x < S; ++x) {
                float* px = wb + (y * S + x) * 3;
                float r = px[0], g = px[1], b = px[2];
                for (int k = 0; k < 3; ++k) {
                    switch (op[k]) {
                        case 0:
                            if (do_b) { r *= bf; g *= bf; b *= bf; }
                            break;
                        case 1:
                            if (do_c) {
                                r = lmean_c + cfc * (r - lmean_c);
                                g = lmean_c + cfc * (g - lmean_c);
                                b = lmean_c + cfc * (b - lmean_c);
                            }
                            break;
                        default:
                            if (do_s) {
                                const float gray = 0.299f * r + 0.587f * g
                                    + 0.114f * b;
                                r = gray + sf * (r - gray);
                                g = gray + sf * (g - gray);
                                b = gray + sf * (b - gray);
                            }
                    }
                }
                if (do_h) {
                    float hh, ss, vv;
                    rgb_to_hsv(clampf(r, 0.f, 255.f) / 255.f,
                               clampf(g, 0.f, 255.f) / 255.f,
                               clampf(b, 0.f, 255.f) / 255.f, hh, ss, vv);
                    hsv_to_rgb(hh + hs, ss, vv, r, g, b);
                    r *= 255.f; g *= 255.f; b *= 255.f;
                }
                const size_t o = (size_t)y * S + x;
                dst[o] = (clampf(r, 0.f, 255.f) / 255.f - nm[0]) / ns[0];
                dst[(size_t)S * S + o] =
                    (clampf(g, 0.f, 255.f) / 255.f - nm[1]) / ns[1];
                dst[2 * (size_t)S * S + o] =
                    (clampf(b, 0.f, 255.f) / 255.f - nm[2]) / ns[2];
            }
        }
    });
    return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("warp_jitter_normalize", &warp_jitter_normalize,
          "fused homography warp + color jitter + normalize (CPU, parallel)");
}
