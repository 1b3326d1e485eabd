/* gcn_demo — end-to-end 2-layer GCN on the C++ operator surface (nts.hpp),
 * driving the gfx950 HIP kernels through the C-ABI with libtorch tensors:
 * the host-side shape of toolkits/GCN_EAGER_single.hpp's train loop
 * (per-layer: graph aggregation op + vertex NN op on the tape, then
 * self_backward + Adam), on synthetic data.
 *
 * Also parity-checks the aggregation against a naive in-file CPU loop
 * (independent of oracle/ — plain double accumulation).
 *
 * Usage: gcn_demo [epochs]   (needs a GPU; exits 2 when none present)
 */
#include <torch/torch.h>

#include <cmath>
#include <cstdio>
#include <random>
#include <vector>

#include "nts/nts.hpp"

using namespace nts;

static void stage(const char *what) { fprintf(stderr, "[demo] %s\n", what); }

int main(int argc, char **argv) {
  stage("start");
  if (!torch::cuda::is_available()) {
    fprintf(stderr, "gcn_demo: no GPU\n");
    return 2;
  }
  stage("cuda available");
  const int epochs = argc > 1 ? atoi(argv[1]) : 10;
  const VertexId V = 4000;
  const uint32_t E = 40000;
  const int F0 = 64, F1 = 32, C = 7;
  torch::manual_seed(7);
  torch::Device dev(torch::kCUDA, 0);

  /* synthetic power-law-ish graph + self loops */
  std::mt19937 rng(7);
  std::vector<uint32_t> src, dst;
  std::uniform_int_distribution<uint32_t> uni(0, V - 1);
  for (uint32_t i = 0; i < E; i++) {
    uint32_t s = uni(rng) % (1 + uni(rng) % V);  /* skewed */
    src.push_back(s);
    dst.push_back(uni(rng));
  }
  for (VertexId v = 0; v < V; v++) { src.push_back(v); dst.push_back(v); }
  const uint32_t Etot = src.size();

  /* degrees (clamped >=1) and norm weights, then CSC + CSR */
  std::vector<uint32_t> outd(V, 0), ind(V, 0);
  for (uint32_t e = 0; e < Etot; e++) { outd[src[e]]++; ind[dst[e]]++; }
  for (VertexId v = 0; v < V; v++) { if (!outd[v]) outd[v] = 1; if (!ind[v]) ind[v] = 1; }
  auto wgt = [&](uint32_t s, uint32_t d) {
    return 1.0f / (std::sqrt((float)outd[s]) * std::sqrt((float)ind[d]));
  };
  std::vector<uint32_t> col_off(V + 1, 0), row_off(V + 1, 0);
  for (uint32_t e = 0; e < Etot; e++) { col_off[dst[e] + 1]++; row_off[src[e] + 1]++; }
  for (VertexId v = 0; v < V; v++) { col_off[v + 1] += col_off[v]; row_off[v + 1] += row_off[v]; }
  std::vector<uint32_t> rows(Etot), cols(Etot), cpos = col_off, rpos = row_off;
  std::vector<float> wf(Etot), wb(Etot);
  for (uint32_t e = 0; e < Etot; e++) {
    uint32_t pc = cpos[dst[e]]++, pr = rpos[src[e]]++;
    rows[pc] = src[e]; wf[pc] = wgt(src[e], dst[e]);
    cols[pr] = dst[e]; wb[pr] = wgt(src[e], dst[e]);
  }

  stage("host graph built");
  auto chunk = CSC_segment_pinned::from_host(
      0, V, 0, V, col_off.data(), rows.data(), wf.data(), row_off.data(),
      cols.data(), wb.data(), Etot, dev);
  stage("chunk uploaded");
  PartitionedGraph pg;
  pg.graph_chunks.push_back(&chunk);
  pg.partition_offset = {0, V};
  VertexSubset active{0, V};

  stage("pg ready");
  /* parity check: one aggregation vs naive double loop on CPU */
  {
    NtsVar x = torch::rand({(int64_t)V, 8}, torch::device(dev)) * 2 - 1;
    op::ForwardSingleGPUfuseOp agg(&pg, &active);
    stage("parity forward");
    NtsVar y = agg.forward(x).cpu();
    stage("parity forward done");
    auto xc = x.cpu().contiguous();
    const float *xp = xc.data_ptr<float>();
    const float *yp = y.data_ptr<float>();
    /* harness tolerance: |err| <= 1e-4*|ref| + 1e-5 (north_star 1e-4 rel
     * fp32, absolute floor for cancellation-small outputs) */
    int bad = 0;
    double worst = 0;
    for (VertexId d = 0; d < V; d++)
      for (int j = 0; j < 8; j++) {
        double acc = 0;
        for (uint32_t e = col_off[d]; e < col_off[d + 1]; e++)
          acc += (double)wf[e] * xp[(int64_t)rows[e] * 8 + j];
        double err = std::abs(acc - yp[(int64_t)d * 8 + j]);
        if (err > 1e-4 * std::abs(acc) + 1e-5) bad++;
        double rel = err / (std::abs(acc) + 1e-5);
        if (rel > worst) worst = rel;
      }
    printf("aggregation parity vs naive CPU: worst rel err %.3e, out-of-tol %d\n",
           worst, bad);
    if (bad) { fprintf(stderr, "PARITY FAIL\n"); return 1; }
  }

  /* training: features, labels, weights */
  NtsVar X = torch::rand({(int64_t)V, F0}, torch::device(dev)) * 2 - 1;
  NtsVar labels = torch::randint(C, {(int64_t)V},
                                 torch::device(dev).dtype(torch::kLong));
  NtsVar W0 = torch::empty({F0, F1}, torch::device(dev)).uniform_(-0.1, 0.1)
                  .set_requires_grad(true);
  NtsVar W1 = torch::empty({F1, C}, torch::device(dev)).uniform_(-0.1, 0.1)
                  .set_requires_grad(true);
  torch::optim::Adam opt({W0, W1}, torch::optim::AdamOptions(1e-2));

  NtsContext ctx;
  double first_loss = 0, last_loss = 0;
  for (int ep = 0; ep < epochs; ep++) {
    opt.zero_grad();
    /* layer 0: aggregate then X·W0+relu under autograd */
    NtsVar a0 = ctx.runGraphOp<op::ForwardSingleGPUfuseOp>(&pg, &active, X);
    NtsVar h0 = ctx.runVertexForward(
        [&](NtsVar &in) { return torch::relu(torch::mm(in, W0)); }, a0);
    NtsVar a1 = ctx.runGraphOp<op::ForwardSingleGPUfuseOp>(&pg, &active, h0);
    NtsVar out = ctx.runVertexForward(
        [&](NtsVar &in) {
          return torch::log_softmax(torch::mm(in, W1), 1);
        }, a1);
    NtsVar loss = torch::nll_loss(out, labels);
    ctx.self_backward(loss);
    opt.step();
    last_loss = loss.item<double>();
    if (ep == 0) first_loss = last_loss;
  }
  printf("loss %0.4f -> %0.4f over %d epochs\n", first_loss, last_loss, epochs);
  if (!(last_loss < first_loss)) { fprintf(stderr, "NO LEARNING\n"); return 1; }

  /* GAT 5-op chain through the decomposed surface (the call shape of
   * toolkits/GAT_GPU_DIST.hpp:191-215): per-edge attention scalars ->
   * softmax -> aggregate; softmax rows must sum to 1 per destination. */
  {
    NtsVar s_src = torch::rand({(int64_t)V, 1}, torch::device(dev)) * 2 - 1;
    NtsVar s_dst = torch::rand({(int64_t)V, 1}, torch::device(dev)) * 2 - 1;
    op::DistGPUGetDepNbrOp dep(&pg, &active);
    op::DistGPUScatterSrc ssrc(&pg, &active);
    op::DistGPUScatterDst sdst(&pg, &active);
    op::DistGPUEdgeSoftMax smax(&pg, &active);
    op::DistGPUAggregateDst aggd(&pg, &active);
    NtsVar mirror = dep.forward(s_src);
    NtsVar m1 = ssrc.forward(mirror);
    NtsVar m2 = sdst.forward(s_dst);
    NtsVar e_att = torch::leaky_relu(m1 + m2, 0.2);
    NtsVar a = smax.forward(e_att);
    NtsVar ones_per_dst = aggd.forward(a);   /* sum of softmax per dst */
    auto sums = ones_per_dst.cpu();
    const float *sp = sums.data_ptr<float>();
    int badsm = 0;
    for (VertexId d = 0; d < V; d++) {
      const bool has_edges = col_off[d + 1] > col_off[d];
      if (has_edges && std::abs(sp[d] - 1.0f) > 1e-4f) badsm++;
    }
    NtsVar ag = smax.backward(a);  /* exercise backward path */
    (void)ag;
    printf("gat 5-op chain: softmax rows off-1: %d\n", badsm);
    if (badsm) { fprintf(stderr, "GAT CHAIN FAIL\n"); return 1; }
  }
  printf("gcn_demo ok\n");
  return 0;
}
