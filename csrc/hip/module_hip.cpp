// pybind11 host module for the gfx950 HIP kernels (`kungfu_amd._hip`).
//
// No torch headers and no hipify: tensors cross as raw device pointers
// (tensor.data_ptr()) and HIP streams as integers
// (torch.cuda.current_stream().cuda_stream); the Python wrapper in
// kungfu_amd/ops/hip.py owns dtype/shape checking.
#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

extern "C" {
hipError_t kf_pack(const void *, int, void *, int, void *);
hipError_t kf_bn_stats(const void *, long long, int, void *, void *);
hipError_t kf_bn_finalize(void *, const void *, const void *, void *,
                          void *, void *, void *, void *, void *, long long,
                          int, float, float, void *);
hipError_t kf_bn_fwd(const void *, const void *, void *, const void *,
                     const void *, long long, int, int, void *, void *);
hipError_t kf_bn_bwd_reduce(const void *, const void *, const void *,
                            const void *, const void *, long long, int,
                            void *, void *);
hipError_t kf_bn_fold(void *, int, void *, void *);
hipError_t kf_col_sum(const void *, long long, int, int, void *, void *,
                      void *);
hipError_t kf_bn_bwd_dx(const void *, const void *, const void *,
                        const void *, const void *, const void *,
                        const void *, long long, int, void *, void *,
                        void *);
hipError_t kf_unpack(const void *, int, const void *, float, int, void *);
hipError_t kf_avg_inplace(void *, const void *, float, long long, int,
                          void *);
hipError_t kf_scale(void *, float, long long, int, void *);
hipError_t kf_norm2(const void *, long long, void *, int, void *);
hipError_t kf_dot(const void *, const void *, long long, void *, int,
                  void *);
hipError_t kf_sgd_momentum(void *, const void *, void *, long long, float,
                           float, float, float, int, int, void *);
hipError_t kf_sgd_momentum_master(void *, const void *, void *, void *,
                                  long long, float, float, float, float,
                                  int, int, void *);
hipError_t kf_transform2(void *, const void *, long long, int, int, void *);
hipError_t kf_ln_fwd(const void *, void *, const void *, const void *,
                     void *, void *, long long, int, float, void *);
hipError_t kf_ln_bwd(const void *, const void *, const void *, const void *,
                     const void *, long long, int, void *, void *, void *);
hipError_t kf_ln_fold(void *, int, void *);
}

namespace {

void check(hipError_t e, const char *what)
{
    if (e != hipSuccess) {
        throw std::runtime_error(std::string(what) + ": " +
                                 hipGetErrorString(e));
    }
}

struct ChunkHost {
    const void *src;
    unsigned long long dst;
    unsigned int n;
};

// Precomputed device-side chunk table for fusion pack/unpack of one bucket.
// segments: list of (device_ptr, fused_elem_offset, numel). The table splits
// segments into <=chunk_elems pieces so blocks load-balance (SURVEY §2.6
// item 6: ResNet-50 has 161 tensors from 1 to 2.3M elements).
class FusionPlan {
  public:
    FusionPlan(const std::vector<std::tuple<uintptr_t, uint64_t, uint64_t>>
                   &segments,
               int dtype, unsigned chunk_elems = 1u << 14)
        : dtype_(dtype), chunk_elems_(chunk_elems)
    {
        build(segments);
        cap_ = nchunks_;
        if (nchunks_ > 0) {
            check(hipMalloc(&dev_, sizeof(ChunkHost) * nchunks_),
                  "hipMalloc(FusionPlan)");
            check(hipHostMalloc(&host_, sizeof(ChunkHost) * nchunks_, 0),
                  "hipHostMalloc(FusionPlan)");
            std::memcpy(host_, chunks_.data(),
                        sizeof(ChunkHost) * nchunks_);
            check(hipMemcpy(dev_, chunks_.data(),
                            sizeof(ChunkHost) * nchunks_,
                            hipMemcpyHostToDevice),
                  "hipMemcpy(FusionPlan)");
        }
    }
    ~FusionPlan()
    {
        if (dev_) (void)hipFree(dev_);
        if (host_) (void)hipHostFree(host_);
    }
    FusionPlan(const FusionPlan &) = delete;

    // Re-point the chunk table at new source buffers (same segment
    // lengths/offsets layout): async H2D from pinned staging on `stream`.
    // Used by the pack-mode gradient path, where autograd's grad tensors
    // get fresh addresses every step.
    void update(const std::vector<std::tuple<uintptr_t, uint64_t,
                                             uint64_t>> &segments,
                uintptr_t stream)
    {
        build(segments);
        if (nchunks_ > cap_)
            throw std::runtime_error("FusionPlan.update: table grew");
        std::memcpy(host_, chunks_.data(), sizeof(ChunkHost) * nchunks_);
        check(hipMemcpyAsync(dev_, host_, sizeof(ChunkHost) * nchunks_,
                             hipMemcpyHostToDevice, (hipStream_t)stream),
              "hipMemcpyAsync(FusionPlan.update)");
    }

    void pack(uintptr_t fused, uintptr_t stream)
    {
        check(kf_pack(dev_, nchunks_, (void *)fused, dtype_,
                      (void *)stream),
              "kf_pack");
    }
    void unpack(uintptr_t fused, float scale, uintptr_t stream)
    {
        check(kf_unpack(dev_, nchunks_, (const void *)fused, scale, dtype_,
                        (void *)stream),
              "kf_unpack");
    }
    uint64_t total_elems() const { return total_; }
    int nchunks() const { return nchunks_; }

  private:
    void build(const std::vector<std::tuple<uintptr_t, uint64_t,
                                            uint64_t>> &segments)
    {
        chunks_.clear();
        total_ = 0;
        for (const auto &seg : segments) {
            const uintptr_t ptr = std::get<0>(seg);
            const uint64_t off = std::get<1>(seg);
            uint64_t n = std::get<2>(seg);
            total_ += n;
            uint64_t done = 0;
            const size_t esize = dtype_ == 0 ? 4 : 2;
            while (done < n) {
                const uint64_t take =
                    std::min<uint64_t>(chunk_elems_, n - done);
                ChunkHost c;
                c.src = (const void *)(ptr + done * esize);
                c.dst = off + done;
                c.n = (unsigned int)take;
                chunks_.push_back(c);
                done += take;
            }
        }
        nchunks_ = (int)chunks_.size();
    }

    std::vector<ChunkHost> chunks_;
    unsigned chunk_elems_;
    int cap_ = 0;
    void *host_ = nullptr;
    int dtype_;
    int nchunks_ = 0;
    void *dev_ = nullptr;
    uint64_t total_ = 0;
};

}  // namespace

PYBIND11_MODULE(_hip, m)
{
    m.doc() = "KungFu-AMD gfx950 HIP kernels";
    m.attr("KERNEL_ARCH") = "gfx950";

    py::class_<FusionPlan>(m, "FusionPlan")
        .def(py::init<const std::vector<
                          std::tuple<uintptr_t, uint64_t, uint64_t>> &,
                      int, unsigned>(),
             py::arg("segments"), py::arg("dtype"),
             py::arg("chunk_elems") = 1u << 14)
        .def("pack", &FusionPlan::pack, py::arg("fused"), py::arg("stream"))
        .def("update", &FusionPlan::update, py::arg("segments"),
             py::arg("stream"))
        .def("unpack", &FusionPlan::unpack, py::arg("fused"),
             py::arg("scale"), py::arg("stream"))
        .def_property_readonly("total_elems", &FusionPlan::total_elems)
        .def_property_readonly("nchunks", &FusionPlan::nchunks);

    m.def("avg_inplace",
          [](uintptr_t y, uintptr_t x, float alpha, long long n, int dtype,
             uintptr_t stream) {
              check(kf_avg_inplace((void *)y, (const void *)x, alpha, n,
                                   dtype, (void *)stream),
                    "kf_avg_inplace");
          });
    m.def("scale_", [](uintptr_t y, float s, long long n, int dtype,
                       uintptr_t stream) {
        check(kf_scale((void *)y, s, n, dtype, (void *)stream), "kf_scale");
    });
    m.def("norm2",
          [](uintptr_t x, long long n, uintptr_t out_f32, int dtype,
             uintptr_t stream) {
              check(kf_norm2((const void *)x, n, (void *)out_f32, dtype,
                             (void *)stream),
                    "kf_norm2");
          });
    m.def("dot",
          [](uintptr_t x, uintptr_t y, long long n, uintptr_t out_f32,
             int dtype, uintptr_t stream) {
              check(kf_dot((const void *)x, (const void *)y, n,
                           (void *)out_f32, dtype, (void *)stream),
                    "kf_dot");
          });
    m.def("sgd_momentum",
          [](uintptr_t p, uintptr_t g, uintptr_t m_f32, long long n,
             float lr, float momentum, float weight_decay, float grad_scale,
             bool nesterov, int dtype, uintptr_t stream) {
              check(kf_sgd_momentum((void *)p, (const void *)g,
                                    (void *)m_f32, n, lr, momentum,
                                    weight_decay, grad_scale,
                                    nesterov ? 1 : 0, dtype,
                                    (void *)stream),
                    "kf_sgd_momentum");
          });
    m.def("sgd_momentum_master",
          [](uintptr_t p, uintptr_t g, uintptr_t master_f32,
             uintptr_t m_f32, long long n, float lr, float momentum,
             float weight_decay, float grad_scale, bool nesterov, int dtype,
             uintptr_t stream) {
              check(kf_sgd_momentum_master(
                        (void *)p, (const void *)g, (void *)master_f32,
                        (void *)m_f32, n, lr, momentum, weight_decay,
                        grad_scale, nesterov ? 1 : 0, dtype,
                        (void *)stream),
                    "kf_sgd_momentum_master");
          });
    m.def("transform2",
          [](uintptr_t z, uintptr_t x, long long n, int op, int dtype,
             uintptr_t stream) {
              check(kf_transform2((void *)z, (const void *)x, n, op, dtype,
                                  (void *)stream),
                    "kf_transform2");
          });
    m.def("device_synchronize", [] { check(hipDeviceSynchronize(), "sync"); });

    // ---- fused LayerNorm (bf16 activations, fp32 params) ----
    m.def("ln_fwd",
          [](uintptr_t x, uintptr_t y, uintptr_t w, uintptr_t b,
             uintptr_t smean, uintptr_t srstd, long long N, int H,
             float eps, uintptr_t stream) {
              check(kf_ln_fwd((const void *)x, (void *)y, (const void *)w,
                              (const void *)b, (void *)smean,
                              (void *)srstd, N, H, eps, (void *)stream),
                    "kf_ln_fwd");
          });
    m.def("ln_bwd",
          [](uintptr_t dy, uintptr_t x, uintptr_t w, uintptr_t smean,
             uintptr_t srstd, long long N, int H, uintptr_t dx,
             uintptr_t wb_sums, uintptr_t stream) {
              check(kf_ln_bwd((const void *)dy, (const void *)x,
                              (const void *)w, (const void *)smean,
                              (const void *)srstd, N, H, (void *)dx,
                              (void *)wb_sums, (void *)stream),
                    "kf_ln_bwd");
          });
    m.def("ln_fold", [](uintptr_t wb, int H, uintptr_t stream) {
        check(kf_ln_fold((void *)wb, H, (void *)stream), "kf_ln_fold");
    });

    // ---- fused BatchNorm(+residual+ReLU), NHWC bf16 ----
    m.def("bn_stats",
          [](uintptr_t x, long long M, int C, uintptr_t sums,
             uintptr_t stream) {
              check(kf_bn_stats((const void *)x, M, C, (void *)sums,
                                (void *)stream),
                    "kf_bn_stats");
          });
    m.def("bn_finalize",
          [](uintptr_t sums, uintptr_t w, uintptr_t bias, uintptr_t rmean,
             uintptr_t rvar, uintptr_t smean, uintptr_t srstd, uintptr_t a,
             uintptr_t b, long long M, int C, float eps, float momentum,
             uintptr_t stream) {
              check(kf_bn_finalize(
                        (void *)sums, (const void *)w,
                        (const void *)bias, (void *)rmean, (void *)rvar,
                        (void *)smean, (void *)srstd, (void *)a, (void *)b,
                        M, C, eps, momentum, (void *)stream),
                    "kf_bn_finalize");
          });
    m.def("bn_fwd",
          [](uintptr_t x, uintptr_t res, uintptr_t y, uintptr_t a,
             uintptr_t b, long long M, int C, bool relu, uintptr_t mask,
             uintptr_t stream) {
              check(kf_bn_fwd((const void *)x, (const void *)res,
                              (void *)y, (const void *)a, (const void *)b,
                              M, C, relu ? 1 : 0, (void *)mask,
                              (void *)stream),
                    "kf_bn_fwd");
          });
    m.def("bn_bwd_reduce",
          [](uintptr_t dy, uintptr_t x, uintptr_t mask, uintptr_t mean,
             uintptr_t rstd, long long M, int C, uintptr_t sums,
             uintptr_t stream) {
              check(kf_bn_bwd_reduce((const void *)dy, (const void *)x,
                                     (const void *)mask,
                                     (const void *)mean,
                                     (const void *)rstd, M, C,
                                     (void *)sums, (void *)stream),
                    "kf_bn_bwd_reduce");
          });
    m.def("bn_fold",
          [](uintptr_t sums, int C, uintptr_t dbdw, uintptr_t stream) {
              check(kf_bn_fold((void *)sums, C, (void *)dbdw,
                               (void *)stream),
                    "kf_bn_fold");
          });
    m.def("col_sum",
          [](uintptr_t dy, long long M, int C, int ld, uintptr_t shadows,
             uintptr_t db, uintptr_t stream) {
              check(kf_col_sum((const void *)dy, M, C, ld,
                               (void *)shadows, (void *)db,
                               (void *)stream),
                    "kf_col_sum");
          });
    m.def("bn_bwd_dx",
          [](uintptr_t dy, uintptr_t x, uintptr_t mask, uintptr_t a,
             uintptr_t mean, uintptr_t rstd, uintptr_t sums, long long M,
             int C, uintptr_t dx, uintptr_t dres, uintptr_t stream) {
              check(kf_bn_bwd_dx((const void *)dy, (const void *)x,
                                 (const void *)mask, (const void *)a,
                                 (const void *)mean, (const void *)rstd,
                                 (const void *)sums, M, C, (void *)dx,
                                 (void *)dres, (void *)stream),
                    "kf_bn_bwd_dx");
          });
}
