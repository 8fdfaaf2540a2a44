"""Serve an OPT model with auto-sharded TP + HTTP frontend (the reference's
examples/llm_serving workflow; random-init weights — no network access).

  torchrun --standalone --nproc-per-node N examples/serve_opt.py \
      --model 1.3B --port 8265
Rank 0 serves HTTP; all ranks participate in TP generation.
"""
import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import alpa_amd as aa
from alpa_amd.models.opt import OPTModel, opt_config
from alpa_amd.serve import (Controller, SpmdGenerateService,
                            run_controller)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="125M")
    p.add_argument("--port", type=int, default=8265)
    p.add_argument("--demo", action="store_true",
                   help="run one generation and exit (no HTTP server)")
    p.add_argument("--fp8", action="store_true",
                   help="e4m3 prefill GEMMs + fp8-packed decode weights "
                        "(measured: 1.58x TTFT, OPT-66B decode +12%%)")
    args = p.parse_args()

    aa.init()
    if args.fp8:
        from alpa_amd.global_env import global_config
        global_config.fp8_gemm = True
    mesh = aa.full_mesh((1, aa.world_size()))
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    cfg = opt_config(args.model, max_seq_len=512)
    model = OPTModel(cfg, mesh, 1, dtype, aa.device(), init_seed=0)

    if args.demo or aa.world_size() > 1:
        ids = torch.randint(0, cfg.vocab_size, (1, 8), device=aa.device())
        out = model.generate(ids, max_new_tokens=8)
        beam = model.beam_search(ids, max_new_tokens=8, num_beams=4)
        if aa.rank() == 0:
            print("generated:", out.tolist())
            print("beam4    :", beam.tolist())
        if args.demo:
            aa.shutdown()
            return
    c = Controller()

    def _gen(ids, mt, num_beams=1):
        ids = ids.to(aa.device())
        if num_beams > 1:
            return model.beam_search(ids, mt, num_beams=num_beams)
        return model.generate(ids, mt)

    svc = SpmdGenerateService(_gen)
    if aa.rank() != 0:
        # non-zero TP ranks execute the driver's broadcast requests
        svc.serve_worker_loop()
        aa.shutdown()
        return

    c.register_model(f"opt-{args.model}", svc)
    run_controller(c, port=args.port)


if __name__ == "__main__":
    main()
