// Python bindings for the cuvite_amd HIP kernels (torch extension).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>
#include <vector>

namespace cuvite {

template <typename W>
struct MoveArgs {
  const int64_t* rowptr;
  const int32_t* tails;
  const W* weights;
  const int32_t* curr_comm;
  const W* v_degree;
  const int64_t* comm_size;
  const W* comm_degree;
  const int64_t* comm_gid;
  double constant;
  int32_t* target;
  W* cluster_weight;
};

template <typename W, int LANES, int CAP>
void launch_sub(const int32_t*, int, const MoveArgs<W>&, hipStream_t);
template <typename W, int CAP>
void launch_block(const int32_t*, int, const MoveArgs<W>&, hipStream_t);
template <typename W>
void launch_modularity(const W*, const W*, int64_t, double*, hipStream_t);
template <typename W>
void launch_scatter_add(W*, const int64_t*, const W*, int64_t, hipStream_t);
void launch_degree_count(const int64_t*, int64_t, int64_t, int32_t*,
                         hipStream_t);
template <typename W>
void launch_csr_place(const int64_t*, const int64_t*, const W*, int64_t,
                      int64_t, const int64_t*, int32_t*, int64_t*, W*,
                      hipStream_t);
template <typename W>
void launch_row_sum(const int64_t*, const W*, int64_t, W*, hipStream_t);
void launch_gather_comm(const int32_t*, const int32_t*, int64_t, int32_t*,
                        hipStream_t);
void launch_pack_key64(const int32_t*, const int32_t*, int64_t, int64_t,
                       int64_t*, hipStream_t);
template <typename W>
void segsort_pairs(void*, size_t*, const int32_t*, int32_t*, const W*, W*,
                   int64_t, int, const int64_t*, int, hipStream_t);
template <typename W>
void reduce_by_key64(void*, size_t*, const int64_t*, const W*, int64_t,
                     int64_t*, W*, unsigned int*, hipStream_t);
template <typename W>
void sort_pairs64(void*, size_t*, const int64_t*, int64_t*, const W*, W*,
                  int64_t, int, hipStream_t);
template <typename W>
void launch_hub_argmax(const int64_t*, const W*, const int32_t*, int64_t,
                       const int32_t*, int, const double*, const int32_t*,
                       const W*, const int64_t*, const W*, const int64_t*,
                       double, double*, int64_t*, int32_t*, int32_t*, W*,
                       hipStream_t);
int hub_argmax_splits();
template <typename W>
void launch_apply_deltas(const int64_t*, const int64_t*, const W*, int64_t,
                         int64_t, int64_t, int64_t*, W*, hipStream_t);
template <typename W>
void launch_recount(const int64_t*, const W*, int64_t, int64_t, int64_t,
                    int64_t*, W*, hipStream_t);
void launch_coloring_minmax(const int64_t*, const int32_t*, const int64_t*,
                            const bool*, const int64_t*, int64_t, int64_t,
                            const int64_t*, int, int64_t*, int64_t*,
                            hipStream_t);

}  // namespace cuvite

namespace {

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

template <typename W>
cuvite::MoveArgs<W> make_args(const at::Tensor& rowptr, const at::Tensor& tails,
                              const at::Tensor& weights,
                              const at::Tensor& curr_comm,
                              const at::Tensor& v_degree,
                              const at::Tensor& comm_size,
                              const at::Tensor& comm_degree,
                              const at::Tensor& comm_gid, double constant,
                              at::Tensor& target, at::Tensor& cw) {
  return cuvite::MoveArgs<W>{
      rowptr.data_ptr<int64_t>(),   tails.data_ptr<int32_t>(),
      weights.data_ptr<W>(),        curr_comm.data_ptr<int32_t>(),
      v_degree.data_ptr<W>(),       comm_size.data_ptr<int64_t>(),
      comm_degree.data_ptr<W>(),    comm_gid.data_ptr<int64_t>(),
      constant,                     target.data_ptr<int32_t>(),
      cw.data_ptr<W>()};
}

// vlists: the 7 LDS degree-class vertex lists (hubs excluded)
// (int32, padded is NOT required; launchers pad logically by bounds checks in
// the sub kernels via nlist).
std::vector<at::Tensor> local_move(
    at::Tensor rowptr, at::Tensor tails, at::Tensor weights,
    at::Tensor curr_comm, at::Tensor v_degree, at::Tensor comm_size,
    at::Tensor comm_degree, at::Tensor comm_gid, double constant,
    std::vector<at::Tensor> vlists) {
  CHECK_DEV(rowptr); CHECK_CONT(rowptr);
  CHECK_DEV(tails); CHECK_CONT(tails);
  CHECK_DEV(weights); CHECK_CONT(weights);
  CHECK_DEV(curr_comm); CHECK_CONT(curr_comm);
  TORCH_CHECK(tails.scalar_type() == at::kInt, "tails must be int32");
  TORCH_CHECK(curr_comm.scalar_type() == at::kInt, "curr_comm must be int32");
  TORCH_CHECK(vlists.size() == 7, "expected 7 degree-class vertex lists"
              " (hub vertices go through the hub_moves pipeline)");

  const int64_t nv = rowptr.numel() - 1;
  auto target = curr_comm.narrow(0, 0, nv).clone();
  auto cw = at::zeros({nv}, weights.options());
  auto stream = at::hip::getCurrentHIPStream().stream();

  AT_DISPATCH_FLOATING_TYPES(weights.scalar_type(), "local_move", [&] {
    using W = scalar_t;
    auto args = make_args<W>(rowptr, tails, weights, curr_comm, v_degree,
                             comm_size, comm_degree, comm_gid, constant,
                             target, cw);
    if (vlists[0].numel())
      cuvite::launch_sub<W, 16, 32>(vlists[0].data_ptr<int32_t>(),
                                    (int)vlists[0].numel(), args, stream);
    if (vlists[1].numel())
      cuvite::launch_sub<W, 64, 128>(vlists[1].data_ptr<int32_t>(),
                                     (int)vlists[1].numel(), args, stream);
    if (vlists[2].numel())
      cuvite::launch_sub<W, 64, 512>(vlists[2].data_ptr<int32_t>(),
                                     (int)vlists[2].numel(), args, stream);
    if (vlists[3].numel())
      cuvite::launch_sub<W, 64, 1024>(vlists[3].data_ptr<int32_t>(),
                                      (int)vlists[3].numel(), args, stream);
    if (vlists[4].numel())
      cuvite::launch_block<W, 2048>(vlists[4].data_ptr<int32_t>(),
                                    (int)vlists[4].numel(), args, stream);
    if (vlists[5].numel())
      cuvite::launch_block<W, 4096>(vlists[5].data_ptr<int32_t>(),
                                    (int)vlists[5].numel(), args, stream);
    if (vlists[6].numel())
      cuvite::launch_block<W, 8192>(vlists[6].data_ptr<int32_t>(),
                                    (int)vlists[6].numel(), args, stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return {target, cw};
}

at::Tensor modularity_parts(at::Tensor cluster_weight, at::Tensor comm_degree) {
  CHECK_DEV(cluster_weight); CHECK_CONT(cluster_weight);
  CHECK_DEV(comm_degree); CHECK_CONT(comm_degree);
  TORCH_CHECK(cluster_weight.numel() == comm_degree.numel());
  auto out = at::zeros({2}, cluster_weight.options().dtype(at::kDouble));
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_FLOATING_TYPES(cluster_weight.scalar_type(), "modularity", [&] {
    cuvite::launch_modularity<scalar_t>(cluster_weight.data_ptr<scalar_t>(),
                                        comm_degree.data_ptr<scalar_t>(),
                                        cluster_weight.numel(),
                                        out.data_ptr<double>(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

void scatter_add_(at::Tensor out, at::Tensor idx, at::Tensor val) {
  CHECK_DEV(out); CHECK_CONT(out);
  CHECK_DEV(idx); CHECK_CONT(idx);
  CHECK_DEV(val); CHECK_CONT(val);
  TORCH_CHECK(idx.scalar_type() == at::kLong, "idx must be int64");
  TORCH_CHECK(idx.numel() == val.numel(), "idx/val length mismatch");
  TORCH_CHECK(out.scalar_type() == val.scalar_type(), "dtype mismatch");
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_FLOATING_TYPES(out.scalar_type(), "scatter_add", [&] {
    cuvite::launch_scatter_add<scalar_t>(out.data_ptr<scalar_t>(),
                                         idx.data_ptr<int64_t>(),
                                         val.data_ptr<scalar_t>(),
                                         idx.numel(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
}

std::vector<at::Tensor> csr_from_edges(int64_t nv, int64_t base,
                                       at::Tensor src, at::Tensor dst,
                                       at::Tensor w) {
  CHECK_DEV(src); CHECK_CONT(src);
  CHECK_DEV(dst); CHECK_CONT(dst);
  CHECK_DEV(w); CHECK_CONT(w);
  TORCH_CHECK(src.scalar_type() == at::kLong && dst.scalar_type() == at::kLong,
              "src/dst must be int64");
  const int64_t ne = src.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto cnt = at::zeros({nv}, src.options().dtype(at::kInt));
  cuvite::launch_degree_count(src.data_ptr<int64_t>(), ne, base,
                              cnt.data_ptr<int32_t>(), stream);
  auto rowptr = at::zeros({nv + 1}, src.options());
  // int64 accumulation: an int32 cumsum overflows past 2^31 total edges
  rowptr.narrow(0, 1, nv).copy_(at::cumsum(cnt, 0, at::kLong));
  auto tails = at::empty({ne}, src.options());
  auto weights = at::empty({ne}, w.options());
  cnt.zero_();  // reuse as the per-row placement cursor
  AT_DISPATCH_FLOATING_TYPES(w.scalar_type(), "csr_place", [&] {
    cuvite::launch_csr_place<scalar_t>(
        src.data_ptr<int64_t>(), dst.data_ptr<int64_t>(),
        w.data_ptr<scalar_t>(), ne, base, rowptr.data_ptr<int64_t>(),
        cnt.data_ptr<int32_t>(), tails.data_ptr<int64_t>(),
        weights.data_ptr<scalar_t>(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return {rowptr, tails, weights};
}

// Full hub move: gather -> segmented sort -> reduce_by_key -> wave-per-hub
// argmax, all device-side (no host sync). Returns per-hub (target, wcc).
std::vector<at::Tensor> hub_moves(
    at::Tensor tails_flat, at::Tensor weights_flat, at::Tensor seg_flat,
    at::Tensor eoffs, at::Tensor hubs_i32, at::Tensor hub_self,
    at::Tensor curr_comm, at::Tensor v_degree, at::Tensor comm_size,
    at::Tensor comm_degree, at::Tensor comm_gid, double constant) {
  CHECK_DEV(tails_flat); CHECK_CONT(tails_flat);
  CHECK_DEV(hubs_i32); CHECK_CONT(hubs_i32);
  TORCH_CHECK(hubs_i32.scalar_type() == at::kInt);
  TORCH_CHECK(hub_self.scalar_type() == at::kDouble);
  const int64_t n = tails_flat.numel();
  TORCH_CHECK(n < (int64_t)INT32_MAX,
              "hub candidate batch exceeds int32 count range; chunk the hubs");
  const int nhub = (int)hubs_i32.numel();
  const int64_t C = comm_degree.numel();
  auto stream = at::hip::getCurrentHIPStream().stream();
  int end_bit = 1;
  while ((int64_t(1) << end_bit) < C) end_bit++;
  auto target_hub = at::empty({nhub}, hubs_i32.options());
  auto cw_hub = at::empty({nhub}, weights_flat.options());
  AT_DISPATCH_FLOATING_TYPES(weights_flat.scalar_type(), "hub_moves", [&] {
    using W = scalar_t;
    auto keys = at::empty({n}, tails_flat.options());
    cuvite::launch_gather_comm(tails_flat.data_ptr<int32_t>(),
                               curr_comm.data_ptr<int32_t>(), n,
                               keys.data_ptr<int32_t>(), stream);
    auto keys2 = at::empty({n}, tails_flat.options());
    auto vals2 = at::empty({n}, weights_flat.options());
    size_t bytes = 0;
    cuvite::segsort_pairs<W>(nullptr, &bytes, keys.data_ptr<int32_t>(),
                             keys2.data_ptr<int32_t>(),
                             weights_flat.data_ptr<W>(), vals2.data_ptr<W>(),
                             n, nhub, eoffs.data_ptr<int64_t>(), end_bit,
                             stream);
    auto temp = at::empty({(int64_t)bytes},
                          tails_flat.options().dtype(at::kByte));
    cuvite::segsort_pairs<W>(temp.data_ptr(), &bytes,
                             keys.data_ptr<int32_t>(),
                             keys2.data_ptr<int32_t>(),
                             weights_flat.data_ptr<W>(), vals2.data_ptr<W>(),
                             n, nhub, eoffs.data_ptr<int64_t>(), end_bit,
                             stream);
    auto key64 = at::empty({n}, eoffs.options());
    cuvite::launch_pack_key64(keys2.data_ptr<int32_t>(),
                              seg_flat.data_ptr<int32_t>(), n, C,
                              key64.data_ptr<int64_t>(), stream);
    auto uniq = at::empty({n}, eoffs.options());
    auto sums = at::empty({n}, weights_flat.options());
    auto cnt = at::zeros({1}, tails_flat.options());
    size_t bytes2 = 0;
    cuvite::reduce_by_key64<W>(nullptr, &bytes2, key64.data_ptr<int64_t>(),
                               vals2.data_ptr<W>(), n,
                               uniq.data_ptr<int64_t>(), sums.data_ptr<W>(),
                               (unsigned int*)cnt.data_ptr<int32_t>(),
                               stream);
    auto temp2 = at::empty({(int64_t)bytes2},
                           tails_flat.options().dtype(at::kByte));
    cuvite::reduce_by_key64<W>(temp2.data_ptr(), &bytes2,
                               key64.data_ptr<int64_t>(),
                               vals2.data_ptr<W>(), n,
                               uniq.data_ptr<int64_t>(), sums.data_ptr<W>(),
                               (unsigned int*)cnt.data_ptr<int32_t>(),
                               stream);
    const int splits = cuvite::hub_argmax_splits();
    auto p_gain = at::empty({(int64_t)nhub * splits},
                            eoffs.options().dtype(at::kDouble));
    auto p_gid = at::empty({(int64_t)nhub * splits},
                           eoffs.options().dtype(at::kLong));
    auto p_dense = at::empty({(int64_t)nhub * splits},
                             tails_flat.options());
    cuvite::launch_hub_argmax<W>(
        uniq.data_ptr<int64_t>(), sums.data_ptr<W>(),
        cnt.data_ptr<int32_t>(), C, hubs_i32.data_ptr<int32_t>(), nhub,
        hub_self.data_ptr<double>(), curr_comm.data_ptr<int32_t>(),
        v_degree.data_ptr<W>(), comm_size.data_ptr<int64_t>(),
        comm_degree.data_ptr<W>(), comm_gid.data_ptr<int64_t>(), constant,
        p_gain.data_ptr<double>(), p_gid.data_ptr<int64_t>(),
        p_dense.data_ptr<int32_t>(),
        target_hub.data_ptr<int32_t>(), cw_hub.data_ptr<W>(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return {target_hub, cw_hub};
}

// Fresh world-1 recount of the community aggregates (see recount_kernel).
void recount_(at::Tensor labels, at::Tensor v_degree, int64_t base,
              at::Tensor size, at::Tensor degree) {
  CHECK_DEV(labels); CHECK_CONT(labels);
  CHECK_DEV(v_degree); CHECK_CONT(v_degree);
  CHECK_DEV(size); CHECK_CONT(size);
  CHECK_DEV(degree); CHECK_CONT(degree);
  TORCH_CHECK(labels.scalar_type() == at::kLong);
  TORCH_CHECK(size.scalar_type() == at::kLong);
  TORCH_CHECK(labels.numel() == v_degree.numel());
  size.zero_();
  degree.zero_();
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_FLOATING_TYPES(degree.scalar_type(), "recount", [&] {
    cuvite::launch_recount<scalar_t>(
        labels.data_ptr<int64_t>(), v_degree.data_ptr<scalar_t>(),
        labels.numel(), base, size.numel(), size.data_ptr<int64_t>(),
        degree.data_ptr<scalar_t>(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
}

// Coarsening aggregate: sort packed (src,dst) keys with weights (narrow-bit
// radix) and sum duplicate keys. Returns (uniq int64 [n], sums W [n],
// count int32[1]) — caller trims to count (ref fill_newEdgesMap +
// duplicate-edge merge, rebuild.cpp:244-279, 379-411, done with nested
// std::maps on the host there).
std::vector<at::Tensor> sort_reduce_pairs(at::Tensor key, at::Tensor val,
                                          int64_t end_bit) {
  CHECK_DEV(key); CHECK_CONT(key);
  CHECK_DEV(val); CHECK_CONT(val);
  TORCH_CHECK(key.scalar_type() == at::kLong, "key must be int64");
  TORCH_CHECK(key.numel() == val.numel());
  const int64_t n = key.numel();
  TORCH_CHECK(n < (int64_t)INT32_MAX, "chunk exceeds int32 count range");
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto uniq = at::empty({std::max<int64_t>(n, 1)}, key.options());
  auto sums = at::empty({std::max<int64_t>(n, 1)}, val.options());
  auto cnt = at::zeros({1}, key.options().dtype(at::kInt));
  if (n == 0) return {uniq.narrow(0, 0, 0), sums.narrow(0, 0, 0), cnt};
  AT_DISPATCH_FLOATING_TYPES(val.scalar_type(), "sort_reduce_pairs", [&] {
    using W = scalar_t;
    auto keys2 = at::empty({n}, key.options());
    auto vals2 = at::empty({n}, val.options());
    size_t bytes = 0;
    cuvite::sort_pairs64<W>(nullptr, &bytes, key.data_ptr<int64_t>(),
                            keys2.data_ptr<int64_t>(), val.data_ptr<W>(),
                            vals2.data_ptr<W>(), n, (int)end_bit, stream);
    auto temp = at::empty({(int64_t)bytes}, key.options().dtype(at::kByte));
    cuvite::sort_pairs64<W>(temp.data_ptr(), &bytes, key.data_ptr<int64_t>(),
                            keys2.data_ptr<int64_t>(), val.data_ptr<W>(),
                            vals2.data_ptr<W>(), n, (int)end_bit, stream);
    size_t bytes2 = 0;
    cuvite::reduce_by_key64<W>(nullptr, &bytes2, keys2.data_ptr<int64_t>(),
                               vals2.data_ptr<W>(), n,
                               uniq.data_ptr<int64_t>(), sums.data_ptr<W>(),
                               (unsigned int*)cnt.data_ptr<int32_t>(),
                               stream);
    auto temp2 = at::empty({(int64_t)bytes2},
                           key.options().dtype(at::kByte));
    cuvite::reduce_by_key64<W>(temp2.data_ptr(), &bytes2,
                               keys2.data_ptr<int64_t>(),
                               vals2.data_ptr<W>(), n,
                               uniq.data_ptr<int64_t>(), sums.data_ptr<W>(),
                               (unsigned int*)cnt.data_ptr<int32_t>(),
                               stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return {uniq, sums, cnt};
}

void apply_deltas_(at::Tensor target, at::Tensor curr, at::Tensor v_degree,
                   int64_t base, int64_t bound, at::Tensor local_size,
                   at::Tensor local_degree) {
  CHECK_DEV(target); CHECK_CONT(target);
  CHECK_DEV(curr); CHECK_CONT(curr);
  CHECK_DEV(v_degree); CHECK_CONT(v_degree);
  CHECK_DEV(local_size); CHECK_CONT(local_size);
  CHECK_DEV(local_degree); CHECK_CONT(local_degree);
  TORCH_CHECK(target.scalar_type() == at::kLong &&
              curr.scalar_type() == at::kLong, "labels must be int64");
  TORCH_CHECK(local_size.scalar_type() == at::kLong);
  TORCH_CHECK(target.numel() == curr.numel() &&
              target.numel() == v_degree.numel());
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_FLOATING_TYPES(local_degree.scalar_type(), "apply_deltas", [&] {
    cuvite::launch_apply_deltas<scalar_t>(
        target.data_ptr<int64_t>(), curr.data_ptr<int64_t>(),
        v_degree.data_ptr<scalar_t>(), target.numel(), base, bound,
        local_size.data_ptr<int64_t>(), local_degree.data_ptr<scalar_t>(),
        stream);
  });
  C10_HIP_CHECK(hipGetLastError());
}

// One coloring round's strict min/max of each hash over competing
// neighbors, per vertex (ref distColoringIteration, coloring.cpp:87-202).
std::vector<at::Tensor> coloring_minmax(at::Tensor rowptr, at::Tensor tails,
                                        at::Tensor gid_all,
                                        at::Tensor uncolored,
                                        at::Tensor colors, int64_t base,
                                        at::Tensor seeds) {
  CHECK_DEV(rowptr); CHECK_CONT(rowptr);
  CHECK_DEV(tails); CHECK_CONT(tails);
  CHECK_DEV(gid_all); CHECK_CONT(gid_all);
  CHECK_DEV(uncolored); CHECK_CONT(uncolored);
  CHECK_DEV(colors); CHECK_CONT(colors);
  TORCH_CHECK(tails.scalar_type() == at::kInt);
  TORCH_CHECK(uncolored.scalar_type() == at::kBool);
  TORCH_CHECK(seeds.scalar_type() == at::kLong && seeds.is_cuda());
  const int n_hash = (int)seeds.numel();
  TORCH_CHECK(1 <= n_hash && n_hash <= 8, "n_hash must be in [1,8]");
  const int64_t nv = rowptr.numel() - 1;
  auto mn = at::empty({nv, n_hash}, rowptr.options());
  auto mx = at::empty({nv, n_hash}, rowptr.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  cuvite::launch_coloring_minmax(
      rowptr.data_ptr<int64_t>(), tails.data_ptr<int32_t>(),
      gid_all.data_ptr<int64_t>(), uncolored.data_ptr<bool>(),
      colors.data_ptr<int64_t>(), nv, base, seeds.data_ptr<int64_t>(),
      n_hash, mn.data_ptr<int64_t>(), mx.data_ptr<int64_t>(), stream);
  C10_HIP_CHECK(hipGetLastError());
  return {mn, mx};
}

at::Tensor row_sum(at::Tensor rowptr, at::Tensor weights) {
  CHECK_DEV(rowptr); CHECK_CONT(rowptr);
  CHECK_DEV(weights); CHECK_CONT(weights);
  const int64_t nv = rowptr.numel() - 1;
  auto out = at::empty({nv}, weights.options());
  auto stream = at::hip::getCurrentHIPStream().stream();
  AT_DISPATCH_FLOATING_TYPES(weights.scalar_type(), "row_sum", [&] {
    cuvite::launch_row_sum<scalar_t>(rowptr.data_ptr<int64_t>(),
                                     weights.data_ptr<scalar_t>(), nv,
                                     out.data_ptr<scalar_t>(), stream);
  });
  C10_HIP_CHECK(hipGetLastError());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("local_move_bucketed", &local_move,
        "Louvain local-move iteration (HIP, degree-class routed)");
  m.def("modularity_parts", &modularity_parts,
        "fp64 (sum cw, sum degree^2) reduction (HIP)");
  m.def("scatter_add_", &scatter_add_,
        "out[idx[i]] += val[i] with native fp atomics (HIP)");
  m.def("csr_from_edges", &csr_from_edges,
        "sort-free CSR assembly on device (HIP)");
  m.def("row_sum", &row_sum, "per-row CSR weight sum (HIP)");
  m.def("apply_deltas_", &apply_deltas_,
        "fused community size/degree delta update for moved vertices (HIP)");
  m.def("sort_reduce_pairs", &sort_reduce_pairs,
        "narrow-bit radix sort + reduce_by_key coarse-edge aggregate "
        "(rocPRIM)");
  m.def("coloring_minmax", &coloring_minmax,
        "per-vertex multi-hash min/max over competing neighbors (HIP)");
  m.def("recount_", &recount_,
        "fresh community size/degree recount from labels (HIP)");
  m.def("hub_moves", &hub_moves,
        "full device-side hub move: segsort + reduce_by_key + argmax");
}
