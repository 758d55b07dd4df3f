// Torch-extension bindings for the MI355X isolation-forest kernels.
// Pure host glue (compiled by g++): tensor checks, LDS sizing, dispatch into
// the hipcc-compiled launchers in forest_kernels.hip.

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

namespace ifa {

void launch_bag_gather(bool bf16, const void* X, const int64_t* bag_idx,
                       float* bags, int64_t T, int64_t n, int64_t d,
                       hipStream_t stream);

void launch_build_forest(const float* bags, const int32_t* feat_sub,
                         int32_t* feat, float* value, int32_t* right,
                         int32_t* count, int32_t* ncount, int32_t* depth,
                         const float* leaf_lut, uint64_t seed,
                         int32_t tree_id_offset, int32_t T, int32_t n,
                         int32_t d, int32_t k, int32_t max_nodes,
                         int32_t height_limit, size_t lds, hipStream_t stream);

void launch_build_extended_forest(const float* bags, const int32_t* feat_sub,
                                  int32_t* feat, float* value, int32_t* right,
                                  int32_t* count, int32_t* ncount,
                                  int32_t* hidx, float* hw, double* off64,
                                  int32_t* depth,
                                  const float* leaf_lut, uint64_t seed,
                                  int32_t tree_id_offset, int32_t T, int32_t n,
                                  int32_t d, int32_t k, int32_t nnz,
                                  int32_t max_nodes, int32_t height_limit,
                                  size_t lds, hipStream_t stream);

void launch_score_forest(bool bf16, int rpt, bool rows_lds, bool nodes_lds,
                         int ilp, const void* X, const void* nodes,
                         const int32_t* ncount, float* out, int64_t N,
                         int32_t d, int32_t dpad, int32_t Tpad,
                         int32_t max_nodes, int32_t height_limit, float fT,
                         float c_norm, int finalize, size_t lds, int blocks,
                         hipStream_t stream);

void launch_score_extended_dense(bool bf16, bool rows_lds, bool wlds,
                                 const void* X, const void* nodes,
                                 const float* hw, const int32_t* ncount,
                                 float* out, int64_t N, int32_t d,
                                 int32_t dpad, int32_t T, int32_t max_nodes,
                                 float fT, float c_norm, int finalize,
                                 size_t lds, int blocks, hipStream_t stream);

void launch_score_extended_forest(bool bf16, bool rows_lds, bool hyper_lds,
                                  bool nodes_lds, const void* X,
                                  const void* nodes, const int32_t* hidx,
                                  const float* hw, const int32_t* ncount,
                                  float* out, int64_t N, int32_t d, int32_t T,
                                  int32_t max_nodes, int32_t nnz, float fT,
                                  float c_norm, int finalize, size_t lds,
                                  int blocks, hipStream_t stream);

void launch_score_extended_sparse_v2(bool bf16, int nnz, const void* X,
                                     const void* nodes, const float* values,
                                     const int32_t* hidx, const float* hw,
                                     const int32_t* ncount, float* out,
                                     int64_t N, int32_t d, int32_t dpad,
                                     int32_t T, int32_t max_nodes,
                                     int32_t height_limit, float fT,
                                     float c_norm, int finalize, size_t lds,
                                     int blocks, hipStream_t stream);

void launch_score_extended_dense_v2(bool bf16, int D, const void* X,
                                    const void* nodes, const float* values,
                                    const float* hw, const int32_t* ncount,
                                    float* out, int64_t N, int32_t d,
                                    int32_t T, int32_t max_nodes,
                                    int32_t height_limit, float fT,
                                    float c_norm, int finalize, size_t lds,
                                    int blocks, hipStream_t stream);

void launch_score_extended_dense_v3(int D, bool rpt2, const void* X,
                                    const void* nodes, const float* values,
                                    const uint32_t* hwp,
                                    const int32_t* ncount, float* out,
                                    int64_t N, int32_t d, int32_t T,
                                    int32_t max_nodes, int32_t height_limit,
                                    float fT, float c_norm, int finalize,
                                    size_t lds, int blocks,
                                    hipStream_t stream);

void launch_score_forest_wide(bool bf16, const void* X, const void* nodes,
                              float* out, int64_t N, int32_t d, int32_t T,
                              int64_t max_nodes, int32_t height_limit,
                              float fT, float c_norm, int finalize,
                              int blocks, hipStream_t stream);

void launch_score_extended_wide(bool bf16, const void* X, const void* nodes,
                                const int32_t* hidx, const float* hw,
                                float* out, int64_t N, int32_t d, int32_t T,
                                int64_t max_nodes, int32_t nnz, float fT,
                                float c_norm, int finalize, int blocks,
                                hipStream_t stream);

}  // namespace ifa

namespace {

constexpr size_t kMaxLds = 160 * 1024;

#define CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be a GPU tensor")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

inline size_t align16(size_t x) { return (x + 15) & ~(size_t)15; }

bool is_bf16(const torch::Tensor& t) {
  return t.scalar_type() == torch::kBFloat16;
}

void check_x(const torch::Tensor& X) {
  TORCH_CHECK(X.dim() == 2, "X must be [N, d]");
  TORCH_CHECK(
      X.scalar_type() == torch::kFloat32 || X.scalar_type() == torch::kBFloat16,
      "X must be float32 or bfloat16");
}

}  // namespace

torch::Tensor bag_gather(torch::Tensor X, torch::Tensor bag_idx) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(bag_idx);
  CHECK_CONTIG(bag_idx);
  check_x(X);
  TORCH_CHECK(bag_idx.dim() == 2 && bag_idx.scalar_type() == torch::kInt64,
              "bag_idx must be int64 [T, n]");
  int64_t T = bag_idx.size(0), n = bag_idx.size(1), d = X.size(1);
  auto bags = torch::empty({T, n, d}, X.options().dtype(torch::kFloat32));
  ifa::launch_bag_gather(is_bf16(X), X.data_ptr(),
                         bag_idx.data_ptr<int64_t>(), bags.data_ptr<float>(),
                         T, n, d, current_stream());
  return bags;
}

std::vector<torch::Tensor> build_forest(torch::Tensor bags,
                                        torch::Tensor feat_sub, int64_t seed,
                                        int64_t tree_id_offset,
                                        torch::Tensor leaf_lut,
                                        int64_t max_nodes,
                                        int64_t height_limit) {
  CHECK_CUDA(bags);
  CHECK_CONTIG(bags);
  CHECK_CUDA(feat_sub);
  CHECK_CONTIG(feat_sub);
  CHECK_CUDA(leaf_lut);
  CHECK_CONTIG(leaf_lut);
  TORCH_CHECK(bags.dim() == 3 && bags.scalar_type() == torch::kFloat32);
  TORCH_CHECK(feat_sub.scalar_type() == torch::kInt32);
  int64_t T = bags.size(0), n = bags.size(1), d = bags.size(2);
  int64_t k = feat_sub.size(1);
  TORCH_CHECK(n <= 16384, "GPU build supports maxSamples <= 16384");
  TORCH_CHECK(max_nodes <= 32767, "max_nodes must fit int16 patch ids");

  auto opts = bags.options().dtype(torch::kInt32);
  auto feat = torch::empty({T, max_nodes}, opts);
  auto value = torch::empty({T, max_nodes}, bags.options());
  auto right = torch::full({T, max_nodes}, -1, opts);
  auto count = torch::empty({T, max_nodes}, opts);
  auto ncount = torch::empty({T}, opts);
  auto depth = torch::zeros({T, max_nodes}, opts);

  size_t lds = align16(4 * n + 2 * k) + (height_limit + 8) * 8;
  TORCH_CHECK(lds <= kMaxLds, "build LDS request too large");
  ifa::launch_build_forest(
      bags.data_ptr<float>(), feat_sub.data_ptr<int32_t>(),
      feat.data_ptr<int32_t>(), value.data_ptr<float>(),
      right.data_ptr<int32_t>(), count.data_ptr<int32_t>(),
      ncount.data_ptr<int32_t>(), depth.data_ptr<int32_t>(),
      leaf_lut.data_ptr<float>(), (uint64_t)seed,
      (int32_t)tree_id_offset, (int32_t)T, (int32_t)n, (int32_t)d, (int32_t)k,
      (int32_t)max_nodes, (int32_t)height_limit, lds, current_stream());
  return {feat, value, right, count, ncount, depth};
}

std::vector<torch::Tensor> build_extended_forest(
    torch::Tensor bags, torch::Tensor feat_sub, int64_t seed,
    int64_t tree_id_offset, torch::Tensor leaf_lut, int64_t nnz,
    int64_t max_nodes, int64_t height_limit) {
  CHECK_CUDA(bags);
  CHECK_CONTIG(bags);
  CHECK_CUDA(feat_sub);
  CHECK_CONTIG(feat_sub);
  CHECK_CUDA(leaf_lut);
  int64_t T = bags.size(0), n = bags.size(1), d = bags.size(2);
  int64_t k = feat_sub.size(1);
  TORCH_CHECK(n <= 16384, "GPU build supports maxSamples <= 16384");
  TORCH_CHECK(max_nodes <= 32767);
  TORCH_CHECK(nnz >= 1 && nnz <= k);

  auto opts = bags.options().dtype(torch::kInt32);
  auto feat = torch::empty({T, max_nodes}, opts);
  auto value = torch::empty({T, max_nodes}, bags.options());
  auto right = torch::full({T, max_nodes}, -1, opts);
  auto count = torch::empty({T, max_nodes}, opts);
  auto ncount = torch::empty({T}, opts);
  auto hidx = torch::zeros({T, max_nodes, nnz}, opts);
  auto hw = torch::zeros({T, max_nodes, nnz}, bags.options());
  auto off64 =
      torch::zeros({T, max_nodes}, bags.options().dtype(torch::kFloat64));
  auto depth = torch::zeros({T, max_nodes}, opts);

  size_t lds =
      align16(4 * n + 2 * k) + align16(8 * nnz) + (height_limit + 8) * 8 + 32;
  TORCH_CHECK(lds <= kMaxLds, "build LDS request too large");
  ifa::launch_build_extended_forest(
      bags.data_ptr<float>(), feat_sub.data_ptr<int32_t>(),
      feat.data_ptr<int32_t>(), value.data_ptr<float>(),
      right.data_ptr<int32_t>(), count.data_ptr<int32_t>(),
      ncount.data_ptr<int32_t>(), hidx.data_ptr<int32_t>(),
      hw.data_ptr<float>(), off64.data_ptr<double>(),
      depth.data_ptr<int32_t>(),
      leaf_lut.data_ptr<float>(), (uint64_t)seed, (int32_t)tree_id_offset,
      (int32_t)T, (int32_t)n, (int32_t)d, (int32_t)k, (int32_t)nnz,
      (int32_t)max_nodes, (int32_t)height_limit, lds, current_stream());
  return {feat, value, right, count, ncount, hidx, hw, off64, depth};
}

torch::Tensor score_forest(torch::Tensor X, torch::Tensor nodes_packed,
                           torch::Tensor ncount, int64_t num_trees,
                           int64_t height_limit, double c_norm,
                           bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_packed);
  CHECK_CONTIG(nodes_packed);
  CHECK_CUDA(ncount);
  check_x(X);
  TORCH_CHECK(nodes_packed.dim() == 3 && nodes_packed.size(2) == 2 &&
                  nodes_packed.scalar_type() == torch::kInt32,
              "nodes must be packed int32 [Tpad, max_nodes, 2]");
  int64_t N = X.size(0), d = X.size(1);
  int64_t Tpad = nodes_packed.size(0), max_nodes = nodes_packed.size(1);
  TORCH_CHECK(Tpad % 8 == 0, "packed tree count must be padded to 8");
  // staged-tree ILP: 4 (default) or 8; 8 halves the number of barrier-
  // bounded staging phases at the cost of occupancy (A/B via env)
  int ilp = 4;
  if (const char* e = getenv("IFA_SCORE_ILP")) ilp = atoi(e) == 8 ? 8 : 4;
  TORCH_CHECK(d <= 4094, "scoring supports d <= 4094 (12-bit feature field "
                         "plus the leaf sentinel column)");
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;

  const bool bf16 = is_bf16(X);
  const size_t elem = bf16 ? 2 : 4;
  // odd-WORD row stride (all 32 LDS banks hit) with room for the sentinel
  // column at index d
  int64_t dpad = d + 1;
  if (bf16) {
    while (dpad % 4 != 2) ++dpad;  // dpad u16 elems -> dpad/2 words, odd
  } else {
    while (dpad % 2 != 1) ++dpad;  // odd word count
  }
  // deep forests (large maxSamples) overflow LDS node staging: walk nodes
  // from global/L2 instead (NODES_LDS=false) — correct for any tree size
  bool nodes_lds =
      (size_t)2 * max_nodes * 8 + (size_t)256 * dpad * elem <= 150 * 1024;
  // config search over (rpt, ilp): wide-d rows inflate the LDS row tile,
  // so dropping staged-tree ILP to 2 can buy back a resident block
  // (measured: d=128 went 1 block/CU at ILP=4 -> 124M rows/s; ILP=2 at 2
  // blocks is the faster shape). Prefer the candidate with the most
  // resident blocks, breaking ties toward more chains (rpt*ilp).
  int rpt = 2;
  bool rows_lds = true;
  size_t lds = 0;
  {
    struct Cand { int rpt, ilp; };
    const Cand cands[] = {{2, ilp}, {1, ilp}, {2, 2}, {1, 2}};
    int best_blocks = 0, best_chains = 0;
    bool found = false;
    for (const Cand& cnd : cands) {
      size_t nb = nodes_lds ? (size_t)cnd.ilp * max_nodes * 8 : 0;
      size_t l = nb + (size_t)cnd.rpt * 256 * dpad * elem;
      if (l > 150 * 1024) continue;
      int blocks_cu = (int)std::min<size_t>(kMaxLds / std::max(l, (size_t)1),
                                            3);
      int chains = cnd.rpt * cnd.ilp;
      if (blocks_cu > best_blocks ||
          (blocks_cu == best_blocks && chains > best_chains)) {
        best_blocks = blocks_cu;
        best_chains = chains;
        rpt = cnd.rpt;
        ilp = cnd.ilp;
        lds = l;
        found = true;
      }
    }
    if (!found) {  // rows exceed LDS even alone: global-X walk
      rows_lds = false;
      rpt = 1;
      lds = nodes_lds ? (size_t)ilp * max_nodes * 8 : 0;
    }
  }
  if (const char* e = getenv("IFA_SCORE_FORCE_GLOBAL")) {
    if (atoi(e)) {
      rows_lds = false;
      rpt = 1;
      lds = nodes_lds ? (size_t)ilp * max_nodes * 8 : 0;
    }
  }
  int64_t rows_per_block = (int64_t)(rows_lds ? rpt : 1) * 256;
  int blocks = (int)std::min<int64_t>(
      (N + rows_per_block - 1) / rows_per_block, 8192);
  ifa::launch_score_forest(bf16, rpt, rows_lds, nodes_lds, ilp, X.data_ptr(),
                           nodes_packed.data_ptr<int32_t>(),
                           ncount.data_ptr<int32_t>(), out.data_ptr<float>(),
                           N, (int32_t)d, (int32_t)dpad, (int32_t)Tpad,
                           (int32_t)max_nodes, (int32_t)height_limit,
                           (float)num_trees, (float)c_norm, finalize ? 1 : 0,
                           lds, blocks, current_stream());
  return out;
}


torch::Tensor score_extended_forest(torch::Tensor X,
                                    torch::Tensor nodes_packed,
                                    torch::Tensor hidx, torch::Tensor hw,
                                    torch::Tensor ncount, double c_norm,
                                    bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_packed);
  CHECK_CONTIG(nodes_packed);
  CHECK_CUDA(hidx);
  CHECK_CONTIG(hidx);
  CHECK_CUDA(hw);
  CHECK_CONTIG(hw);
  check_x(X);
  TORCH_CHECK(nodes_packed.dim() == 3 && nodes_packed.size(2) == 2 &&
                  nodes_packed.scalar_type() == torch::kInt32,
              "nodes must be packed int32 [T, max_nodes, 2]");
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_packed.size(0), max_nodes = nodes_packed.size(1);
  int64_t nnz = hidx.size(2);
  TORCH_CHECK(d <= 4095, "scoring supports d <= 4095");
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;

  const bool bf16 = is_bf16(X);
  const size_t elem = bf16 ? 2 : 4;
  const size_t pad = 16 / elem;
  size_t row_bytes = (size_t)256 * (d + pad) * elem;
  size_t node_bytes = (size_t)max_nodes * 8;
  int blocks = (int)std::min<int64_t>((N + 255) / 256, 8192);

  if (nnz == d && (d % 4) == 0 && node_bytes <= 120 * 1024) {
    // fully-extended dense hyperplanes: implicit indices, vectorized dot.
    // 8-B-aligned row stride for the uint2/float4 row reads.
    int64_t pad4 = (4 - (d % 4)) % 4;
    int64_t dpad = d + (pad4 == 0 ? 4 : pad4);
    size_t drow_bytes = (size_t)256 * dpad * elem;
    size_t w_bytes = (size_t)max_nodes * d * 4;
    bool wlds = node_bytes + w_bytes + drow_bytes <= 152 * 1024;
    bool rows_lds =
        node_bytes + (wlds ? w_bytes : 0) + drow_bytes <= 152 * 1024;
    size_t lds = node_bytes + (wlds ? w_bytes : 0) + (rows_lds ? drow_bytes : 0);
    ifa::launch_score_extended_dense(
        bf16, rows_lds, wlds, X.data_ptr(), nodes_packed.data_ptr<int32_t>(),
        hw.data_ptr<float>(), ncount.data_ptr<int32_t>(),
        out.data_ptr<float>(), N, (int32_t)d, (int32_t)dpad, (int32_t)T,
        (int32_t)max_nodes, (float)T, (float)c_norm, finalize ? 1 : 0, lds,
        blocks, current_stream());
    return out;
  }

  // deep forests: stage what fits, walk the rest from global/L2
  bool nodes_lds = node_bytes <= 120 * 1024;
  size_t nb = nodes_lds ? node_bytes : 0;
  size_t hyper_bytes = (size_t)max_nodes * nnz * 8;
  bool hyper_lds = nb + hyper_bytes <= 96 * 1024;
  bool rows_lds =
      nb + (hyper_lds ? hyper_bytes : 0) + row_bytes <= 144 * 1024;
  size_t lds =
      nb + (hyper_lds ? hyper_bytes : 0) + (rows_lds ? row_bytes : 0);
  ifa::launch_score_extended_forest(
      bf16, rows_lds, hyper_lds, nodes_lds, X.data_ptr(),
      nodes_packed.data_ptr<int32_t>(), hidx.data_ptr<int32_t>(),
      hw.data_ptr<float>(), ncount.data_ptr<int32_t>(), out.data_ptr<float>(),
      N, (int32_t)d, (int32_t)T, (int32_t)max_nodes, (int32_t)nnz, (float)T,
      (float)c_norm, finalize ? 1 : 0, lds, blocks, current_stream());
  return out;
}

torch::Tensor score_extended_dense_v2(torch::Tensor X,
                                      torch::Tensor nodes_packed,
                                      torch::Tensor values, torch::Tensor hw,
                                      torch::Tensor ncount,
                                      int64_t height_limit, double c_norm,
                                      bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_packed);
  CHECK_CONTIG(nodes_packed);
  CHECK_CUDA(values);
  CHECK_CONTIG(values);
  CHECK_CUDA(hw);
  CHECK_CONTIG(hw);
  CHECK_CUDA(ncount);
  check_x(X);
  TORCH_CHECK(nodes_packed.dim() == 3 && nodes_packed.size(2) == 2 &&
                  nodes_packed.scalar_type() == torch::kInt32,
              "nodes must be packed int32 [T, max_nodes, 2]");
  TORCH_CHECK(hw.dim() == 3 && hw.scalar_type() == torch::kFloat32,
              "hw must be float32 [T, max_nodes, d]");
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_packed.size(0), max_nodes = nodes_packed.size(1);
  TORCH_CHECK(d <= 64, "dense v2 supports d <= 64");
  const int D = d <= 8 ? 8 : (d <= 16 ? 16 : (d <= 32 ? 32 : 64));
  TORCH_CHECK(hw.size(2) == D, "hw must be host-padded to D columns");
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;

  size_t lds = (size_t)max_nodes * 12 + 16 + (size_t)max_nodes * (D / 4 + 1) * 16;
  TORCH_CHECK(lds <= kMaxLds, "tree too large for LDS staging");
  int blocks = (int)std::min<int64_t>((N + 511) / 512, 8192);
  ifa::launch_score_extended_dense_v2(
      is_bf16(X), D, X.data_ptr(), nodes_packed.data_ptr<int32_t>(),
      values.data_ptr<float>(), hw.data_ptr<float>(),
      ncount.data_ptr<int32_t>(), out.data_ptr<float>(), N, (int32_t)d,
      (int32_t)T, (int32_t)max_nodes, (int32_t)height_limit, (float)T,
      (float)c_norm, finalize ? 1 : 0, lds, blocks, current_stream());
  return out;
}

torch::Tensor score_extended_dense_v3(torch::Tensor X,
                                      torch::Tensor nodes_packed,
                                      torch::Tensor values, torch::Tensor hwp,
                                      torch::Tensor ncount,
                                      int64_t height_limit, double c_norm,
                                      bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_packed);
  CHECK_CONTIG(nodes_packed);
  CHECK_CUDA(values);
  CHECK_CONTIG(values);
  CHECK_CUDA(hwp);
  CHECK_CONTIG(hwp);
  CHECK_CUDA(ncount);
  check_x(X);
  TORCH_CHECK(is_bf16(X), "dense v3 requires bf16 rows (f32 rows route to v2)");
  TORCH_CHECK(nodes_packed.dim() == 3 && nodes_packed.size(2) == 2 &&
                  nodes_packed.scalar_type() == torch::kInt32,
              "nodes must be packed int32 [T, max_nodes, 2]");
  TORCH_CHECK(hwp.dim() == 3 && hwp.scalar_type() == torch::kInt32,
              "hwp must be int32 [T, max_nodes, D/2] packed bf16 pairs");
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_packed.size(0), max_nodes = nodes_packed.size(1);
  TORCH_CHECK(d <= 128, "dense v3 supports d <= 128");
  const int D = d <= 8 ? 8
                       : (d <= 16 ? 16
                                  : (d <= 32 ? 32 : (d <= 64 ? 64 : 128)));
  TORCH_CHECK(hwp.size(2) == D / 2, "hwp must be packed to D/2 dwords");
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;

  size_t lds = (size_t)max_nodes * 12 + 16 + (size_t)max_nodes * (D / 8 + 1) * 16;
  TORCH_CHECK(lds <= kMaxLds, "tree too large for LDS staging");
  const char* rpt_env = getenv("IFA_EIF_V3_RPT2");
  const bool rpt2 = rpt_env && rpt_env[0] == '1';
  const int rows_per_iter = (D >= 64 || (D == 32 && !rpt2)) ? 512 : 1024;
  int blocks = (int)std::min<int64_t>(
      (N + rows_per_iter - 1) / rows_per_iter, 8192);
  ifa::launch_score_extended_dense_v3(
      D, rpt2, X.data_ptr(), nodes_packed.data_ptr<int32_t>(),
      values.data_ptr<float>(), (const uint32_t*)hwp.data_ptr<int32_t>(),
      ncount.data_ptr<int32_t>(), out.data_ptr<float>(), N, (int32_t)d,
      (int32_t)T, (int32_t)max_nodes, (int32_t)height_limit, (float)T,
      (float)c_norm, finalize ? 1 : 0, lds, blocks, current_stream());
  return out;
}

torch::Tensor score_extended_sparse_v2(torch::Tensor X,
                                       torch::Tensor nodes_packed,
                                       torch::Tensor values,
                                       torch::Tensor hidx, torch::Tensor hw,
                                       torch::Tensor ncount,
                                       int64_t height_limit, double c_norm,
                                       bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_packed);
  CHECK_CONTIG(nodes_packed);
  CHECK_CUDA(values);
  CHECK_CONTIG(values);
  CHECK_CUDA(hidx);
  CHECK_CONTIG(hidx);
  CHECK_CUDA(hw);
  CHECK_CONTIG(hw);
  CHECK_CUDA(ncount);
  check_x(X);
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_packed.size(0), max_nodes = nodes_packed.size(1);
  int64_t nnz = hidx.size(2);
  TORCH_CHECK(nnz >= 1 && nnz <= 5, "sparse v2 supports nnz <= 5");
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;

  const bool bf16 = is_bf16(X);
  const size_t elem = bf16 ? 2 : 4;
  int64_t dpad = d;
  if (bf16) {
    while (dpad % 4 != 2) ++dpad;
  } else {
    while (dpad % 2 != 1) ++dpad;
  }
  size_t lds = (size_t)max_nodes * (12 + nnz * 8)
               + (size_t)2 * 256 * dpad * elem;
  TORCH_CHECK(lds <= 150 * 1024, "sparse v2 LDS overflow; use general path");
  int blocks = (int)std::min<int64_t>((N + 511) / 512, 8192);
  ifa::launch_score_extended_sparse_v2(
      bf16, (int)nnz, X.data_ptr(), nodes_packed.data_ptr<int32_t>(),
      values.data_ptr<float>(), hidx.data_ptr<int32_t>(),
      hw.data_ptr<float>(), ncount.data_ptr<int32_t>(),
      out.data_ptr<float>(), N, (int32_t)d, (int32_t)dpad, (int32_t)T,
      (int32_t)max_nodes, (int32_t)height_limit, (float)T, (float)c_norm,
      finalize ? 1 : 0, lds, blocks, current_stream());
  return out;
}

torch::Tensor score_forest_wide(torch::Tensor X, torch::Tensor nodes_wide,
                                int64_t height_limit, double c_norm,
                                bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_wide);
  CHECK_CONTIG(nodes_wide);
  check_x(X);
  TORCH_CHECK(nodes_wide.dim() == 3 && nodes_wide.size(2) == 4 &&
                  nodes_wide.scalar_type() == torch::kInt32,
              "nodes must be wide int32 [T, max_nodes, 4]");
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_wide.size(0), max_nodes = nodes_wide.size(1);
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;
  int blocks = (int)std::min<int64_t>((N + 255) / 256, 8192);
  ifa::launch_score_forest_wide(
      is_bf16(X), X.data_ptr(), nodes_wide.data_ptr<int32_t>(),
      out.data_ptr<float>(), N, (int32_t)d, (int32_t)T, max_nodes,
      (int32_t)height_limit, (float)T, (float)c_norm, finalize ? 1 : 0,
      blocks, current_stream());
  return out;
}

torch::Tensor score_extended_wide(torch::Tensor X, torch::Tensor nodes_wide,
                                  torch::Tensor hidx, torch::Tensor hw,
                                  double c_norm, bool finalize) {
  CHECK_CUDA(X);
  CHECK_CONTIG(X);
  CHECK_CUDA(nodes_wide);
  CHECK_CONTIG(nodes_wide);
  CHECK_CUDA(hidx);
  CHECK_CONTIG(hidx);
  CHECK_CUDA(hw);
  CHECK_CONTIG(hw);
  check_x(X);
  TORCH_CHECK(nodes_wide.dim() == 3 && nodes_wide.size(2) == 4 &&
                  nodes_wide.scalar_type() == torch::kInt32,
              "nodes must be wide int32 [T, max_nodes, 4]");
  int64_t N = X.size(0), d = X.size(1);
  int64_t T = nodes_wide.size(0), max_nodes = nodes_wide.size(1);
  int64_t nnz = hidx.size(2);
  auto out = torch::empty({N}, X.options().dtype(torch::kFloat32));
  if (N == 0) return out;
  int blocks = (int)std::min<int64_t>((N + 255) / 256, 8192);
  ifa::launch_score_extended_wide(
      is_bf16(X), X.data_ptr(), nodes_wide.data_ptr<int32_t>(),
      hidx.data_ptr<int32_t>(), hw.data_ptr<float>(), out.data_ptr<float>(),
      N, (int32_t)d, (int32_t)T, max_nodes, (int32_t)nnz, (float)T,
      (float)c_norm, finalize ? 1 : 0, blocks, current_stream());
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bag_gather", &bag_gather, "gather per-tree bags (K9/K10)");
  m.def("build_forest", &build_forest, "build standard iTrees (K1/K2/K11)");
  m.def("build_extended_forest", &build_extended_forest,
        "build extended iTrees (K3-K5)");
  m.def("score_forest", &score_forest, "batched path-length scoring (K6)");
  m.def("score_extended_forest", &score_extended_forest,
        "batched EIF scoring (K7)");
  m.def("score_extended_sparse_v2", &score_extended_sparse_v2,
        "sparse EIF scoring, fixed-trip batched walk (K7 small-nnz path)");
  m.def("score_extended_dense_v2", &score_extended_dense_v2,
        "dense EIF scoring, rows-in-registers (K7 fast path)");
  m.def("score_extended_dense_v3", &score_extended_dense_v3,
        "dense EIF scoring, packed-bf16 weights + v_dot2c (K7 bf16 path)");
  m.def("score_forest_wide", &score_forest_wide,
        "wide-format standard scoring (no packed-field limits)");
  m.def("score_extended_wide", &score_extended_wide,
        "wide-format EIF scoring (no packed-field limits)");
  m.attr("WAVE") = 64;
}
