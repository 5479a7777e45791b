// pybind11 bindings for the hand-written CDNA4 HIP kernels.

#include <torch/extension.h>

std::vector<at::Tensor> fused_bn_relu_forward(
    at::Tensor x, at::Tensor gamma, at::Tensor beta,
    c10::optional<at::Tensor> running_mean,
    c10::optional<at::Tensor> running_var, double eps, double momentum,
    bool fuse_relu);

at::Tensor bn_inference_apply(at::Tensor x, at::Tensor scale,
                              at::Tensor shift, bool fuse_relu);

std::vector<at::Tensor> fused_bn_relu_backward(
    at::Tensor dy, at::Tensor x, at::Tensor gamma, at::Tensor beta,
    at::Tensor stats, bool fused_relu);

at::Tensor mfma_probe(at::Tensor A, at::Tensor B);

at::Tensor conv_s1_nhwc(at::Tensor x, at::Tensor wpk, int64_t K,
                        int64_t R, int64_t S, int64_t pad);
at::Tensor conv_s1_nhwc_cchunk(at::Tensor x, at::Tensor wpk, int64_t K,
                               int64_t R, int64_t S, int64_t pad);
std::vector<at::Tensor> mdn_nll_forward(at::Tensor params,
                                        at::Tensor labels, int64_t A,
                                        int64_t S);
at::Tensor mdn_nll_backward(at::Tensor params, at::Tensor labels,
                            at::Tensor wsave, at::Tensor gout,
                            int64_t A, int64_t S);

at::Tensor pack_conv_w(at::Tensor w, bool transpose);
std::vector<at::Tensor> pack_conv_w_pair(at::Tensor w);
std::vector<at::Tensor> spatial_softmax_fwd(at::Tensor x, double temp);
at::Tensor spatial_softmax_bwd(at::Tensor map, at::Tensor points,
                               c10::optional<at::Tensor> dpoints,
                               c10::optional<at::Tensor> dmap,
                               double temp);
at::Tensor s2d_stem(at::Tensor x);

at::Tensor conv_s1_wrw(at::Tensor x, at::Tensor dy, int64_t R, int64_t S,
                       int64_t pad);
at::Tensor conv_s1_wrw2(at::Tensor x, at::Tensor dy, int64_t R, int64_t S,
                        int64_t pad);
at::Tensor conv_s1_wrw3(at::Tensor x, at::Tensor dy, int64_t R, int64_t S,
                        int64_t pad);
at::Tensor conv_s1_wrw4(at::Tensor x, at::Tensor dy, int64_t R, int64_t S,
                        int64_t pad);
at::Tensor im2col_nhwc(at::Tensor x, int64_t R, int64_t S, int64_t pad,
                       int64_t stride);
at::Tensor col2im_nhwc(at::Tensor dcol, int64_t N, int64_t C, int64_t H,
                       int64_t W, int64_t R, int64_t S, int64_t pad,
                       int64_t stride);
at::Tensor jpeg_idct(at::Tensor coeffs, at::Tensor quant, int64_t bh,
                     int64_t bw);
at::Tensor jpeg_color(at::Tensor py, at::Tensor pcb, at::Tensor pcr,
                      int64_t H, int64_t W, int64_t hs_y, int64_t vs_y,
                      int64_t hs_c, int64_t vs_c, int64_t hmax,
                      int64_t vmax);
at::Tensor jpeg_gray(at::Tensor py, int64_t H, int64_t W);

at::Tensor conv_stem_nhwc(at::Tensor x, at::Tensor wpk);

at::Tensor pack_stem_w(at::Tensor w);

std::vector<at::Tensor> maxpool_nhwc_forward(at::Tensor x, int64_t kh,
                                             int64_t kw, bool ceil_mode);

at::Tensor maxpool_nhwc_backward(at::Tensor dy, at::Tensor argmax,
                                 int64_t N, int64_t C, int64_t H,
                                 int64_t W, int64_t kh, int64_t kw);

at::Tensor fused_preprocess(at::Tensor raw, int64_t oy, int64_t ox,
                            int64_t th, int64_t tw,
                            c10::optional<at::Tensor> delta_b,
                            c10::optional<at::Tensor> f_sat,
                            c10::optional<at::Tensor> f_con, bool out_bf16);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_bn_relu_forward", &fused_bn_relu_forward,
        "Fused training BN(+ReLU) forward (NHWC bf16)");
  m.def("bn_inference_apply", &bn_inference_apply,
        "BN inference apply (+ReLU) (NHWC bf16)");
  m.def("fused_bn_relu_backward", &fused_bn_relu_backward,
        "Fused BN(+ReLU) backward (NHWC bf16)");
  m.def("mfma_probe", &mfma_probe, "32x32x16 bf16 MFMA layout probe");
  m.def("mdn_nll_forward", &mdn_nll_forward,
        "fused MDN negative log-likelihood forward");
  m.def("mdn_nll_backward", &mdn_nll_backward,
        "fused MDN negative log-likelihood backward");
  m.def("conv_s1_nhwc_cchunk", &conv_s1_nhwc_cchunk,
        "C-chunked MFMA 3x3 stride-1 conv (C%32, K%64, NHWC bf16)");
  m.def("conv_s1_nhwc", &conv_s1_nhwc,
        "MFMA stride-1 NHWC bf16 conv (prepacked weights)");
  m.def("pack_conv_w", &pack_conv_w,
        "single-kernel conv weight pack (transpose=bwd-data layout)");
  m.def("pack_conv_w_pair", &pack_conv_w_pair,
        "fwd + dgrad weight packs in one dispatch");
  m.def("s2d_stem", &s2d_stem,
        "fused space-to-depth [N,3,H,W]->[N,16,H/2,W/2] for the stem");
  m.def("conv_s1_wrw4", &conv_s1_wrw4,
        "MFMA wrw v4: pixel-major x image + ds_read_b64_tr_b16 A-reads");
  m.def("im2col_nhwc", &im2col_nhwc,
        "NHWC bf16 im2col gather (GEMM-conv path)");
  m.def("col2im_nhwc", &col2im_nhwc,
        "NHWC bf16 col2im gather (GEMM-conv dgrad)");
  m.def("jpeg_idct", &jpeg_idct,
        "JPEG dequant + batched 8x8 IDCT -> f32 component plane");
  m.def("jpeg_color", &jpeg_color,
        "JPEG upsample + YCbCr->RGB -> uint8 NHWC");
  m.def("jpeg_gray", &jpeg_gray, "JPEG grayscale plane -> uint8");
  m.def("conv_s1_wrw3", &conv_s1_wrw3,
        "MFMA wrw v3: occupancy-first rs-split (256-thr WGs, 3/CU)");
  m.def("conv_s1_wrw2", &conv_s1_wrw2,
        "MFMA wrw v2: register-accumulator dW, single staging pass");
  m.def("conv_s1_wrw", &conv_s1_wrw,
        "MFMA stride-1 conv weight gradient (LDS-accumulated rs groups)");
  m.def("conv_stem_nhwc", &conv_stem_nhwc,
        "MFMA 6x6/2 C=3 stem conv (implicit im2col)");
  m.def("pack_stem_w", &pack_stem_w, "stem weight pack [12][64][24]");
  m.def("spatial_softmax_fwd", &spatial_softmax_fwd,
        "fused online-softmax soft arg-max: points [N,2C] + map");
  m.def("spatial_softmax_bwd", &spatial_softmax_bwd,
        "fused spatial softmax backward");
  m.def("maxpool_nhwc_forward", &maxpool_nhwc_forward,
        "Non-overlapping NHWC bf16 max-pool forward (+argmax)");
  m.def("maxpool_nhwc_backward", &maxpool_nhwc_backward,
        "Non-overlapping NHWC bf16 max-pool backward (gather)");
  m.def("fused_preprocess", &fused_preprocess,
        "Fused crop+convert+photometric distortion (uint8 NHWC -> f32/bf16)");
}
