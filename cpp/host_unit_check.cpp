/* host_unit_check — CPU-runnable unit checks for the C++ host layer
 * (runs in the no-GPU suite, unlike gcn_demo/gcn_link_check whose compute
 * needs a device):
 *
 *  1. nts::Parameter's hand-rolled Adam (learnC2G_with_decay_Adam + next,
 *     NtsScheduler.hpp:639-791 surface) against an independent double-
 *     precision recomputation of the same update rule, including the
 *     reference's running bias-correction and decay bookkeeping.
 *  2. nts::NtsContext's tape (runGraphOp + runVertexForward +
 *     self_backward, ntsContext.hpp:108-359 surface) on CPU tensors with
 *     an analytically-checkable graph op (y = 2x) and NN segment
 *     (z = x.sum()*w): d z / d x must come out as 2*w.
 *
 * Exit 0 on success; prints the failing check otherwise.
 */
#include <torch/torch.h>

#include <cmath>
#include <cstdio>
#include <vector>

#include "nts/nts.hpp"

using namespace nts;

static int fail(const char *what) {
  fprintf(stderr, "host_unit_check FAIL: %s\n", what);
  return 1;
}

/* trivial graph op: forward doubles, backward halves — lets the tape's
 * graph-op/NN-op interleaving be checked analytically */
struct DoubleOp : public op::ntsGraphOp {
  DoubleOp(PartitionedGraph *pg, VertexSubset *a) : ntsGraphOp(pg, a) {}
  NtsVar forward(NtsVar &x) override { return x * 2.0; }
  NtsVar backward(NtsVar &g) override { return g * 2.0; }
};

int main() {
  torch::manual_seed(5);

  /* ---- 1. Parameter Adam vs double recomputation ---- */
  {
    const int R = 4, C = 3;
    Parameter P(R, C, /*alpha*/ 0.01f, 0.9f, 0.999f, 1e-9f,
                /*weight_decay*/ 0.05f);
    P.Adam_to_GPU(torch::kCPU);
    P.set_decay(0.5f, 2);
    std::vector<double> w(R * C), m(R * C, 0), vv(R * C, 0);
    {
      auto a = P.W.detach().contiguous();
      const float *p = a.data_ptr<float>();
      for (int i = 0; i < R * C; i++) w[i] = p[i];
    }
    double alpha = 0.01, b1 = 0.9, b2 = 0.999, eps = 1e-9, wd = 0.05;
    double a_t = 0.01, b1_t = 0.9, b2_t = 0.999;
    long curr_epoch = 0;
    for (int step = 0; step < 5; step++) {
      NtsVar g = torch::rand({R, C}) - 0.5;
      /* model: same order as the reference's Update() loop —
       * all_reduce (P=1: copy), learn, next */
      P.all_reduce_to_gradient(g.clone());
      P.learnC2G_with_decay_Adam();
      P.next();
      const float *gp = g.data_ptr<float>();
      for (int i = 0; i < R * C; i++) {
        double gg = gp[i] + wd * w[i];
        m[i] = b1 * m[i] + (1 - b1) * gg;
        vv[i] = b2 * vv[i] + (1 - b2) * gg * gg;
        w[i] = w[i] - alpha * m[i] / (std::sqrt(vv[i]) + eps);
      }
      /* next() bookkeeping (NtsScheduler.hpp:725-733 exactly) */
      if (curr_epoch != 0 && curr_epoch % 2 == 0) a_t *= 0.5;
      alpha = a_t * std::sqrt(1 - b2) / (1 - b1);
      b1 *= b1_t;
      b2 *= b2_t;
      curr_epoch++;
    }
    auto a = P.W.detach().contiguous();
    const float *p = a.data_ptr<float>();
    for (int i = 0; i < R * C; i++) {
      if (std::abs(p[i] - w[i]) > 1e-4 * std::abs(w[i]) + 1e-5)
        return fail("Parameter Adam diverged from double recomputation");
    }
    /* the class runs the bookkeeping in fp32; the recomputation in double */
    if (std::abs(P.alpha - alpha) > 1e-5 * std::abs(alpha))
      return fail("next() alpha drift");
    printf("Parameter Adam + decay bookkeeping ok\n");
  }

  /* ---- 2. NtsContext tape on CPU ---- */
  {
    PartitionedGraph pg;          /* no chunks needed for DoubleOp */
    pg.partition_offset = {0, 8};
    VertexSubset active{0, 8};
    NtsVar w = torch::tensor({3.0f}).set_requires_grad(true);
    NtsVar x = torch::rand({8, 4});
    NtsContext ctx;
    NtsVar y = ctx.runGraphOp<DoubleOp>(&pg, &active, x);       /* y = 2x */
    NtsVar z = ctx.runVertexForward(
        [&](NtsVar &in) { return in.sum() * w; }, y);           /* z = w·Σy */
    NtsVar gx = ctx.self_backward(z);
    /* dz/dx = 2 * w everywhere */
    if (!torch::allclose(gx, torch::full({8, 4}, 6.0f)))
      return fail("tape gradient != 2*w");
    if (!w.grad().defined() ||
        std::abs(w.grad().item<float>() - (2 * x.sum()).item<float>()) > 1e-3)
      return fail("NN-segment weight grad wrong");
    printf("NtsContext tape (graph op + NN segment) ok\n");
  }

  printf("host_unit_check ok\n");
  return 0;
}
