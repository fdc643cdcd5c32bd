"""REST text-generation server (reference megatron/text_generation_server.py
+ tools/run_text_generation_server.py).

The reference uses Flask; Flask is not in this image, so the same REST
contract (PUT /api with a JSON body of prompts + sampling params, lock-
serialized, rank 0 serves while other ranks loop on a broadcast choice) is
implemented on FastAPI/uvicorn which are available offline. The wire format
is unchanged.
"""

import threading

import torch

from .api import beam_search_and_post_process, generate_and_post_process
from .communication import broadcast_float_list

GENERATE_NUM = 0
BEAM_NUM = 1
LOCK = threading.Lock()


def send_do_generate():
    choice = torch.zeros(1, dtype=torch.float32)
    broadcast_float_list(1, float_list=[GENERATE_NUM])
    return choice


def send_do_beam_search():
    choice = torch.ones(1, dtype=torch.float32)
    broadcast_float_list(1, float_list=[BEAM_NUM])
    return choice


def run_worker_loop(model):
    """Non-rank-0 ranks: wait for the broadcast choice then join the
    collective generate (reference tools/run_text_generation_server.py)."""
    while True:
        choice = broadcast_float_list(1, float_list=None)
        if int(choice[0].item()) == GENERATE_NUM:
            generate_and_post_process(model)
        else:
            beam_search_and_post_process(model)


class MegatronServer:
    def __init__(self, model):
        self.model = model

    def _build_app(self):
        from fastapi import FastAPI, Request
        from fastapi.responses import JSONResponse

        app = FastAPI()
        model = self.model

        @app.put("/api")
        async def api(request: Request):
            raw = await request.json()
            if "prompts" not in raw:
                return JSONResponse({"message": "No prompts supplied"},
                                    status_code=400)
            prompts = raw["prompts"]
            if not isinstance(prompts, list) or len(prompts) == 0:
                return JSONResponse({"message": "prompts is not a nonempty list"},
                                    status_code=400)
            tokens_to_generate = int(raw.get("tokens_to_generate", 64))
            logprobs = bool(raw.get("logprobs", False))
            temperature = float(raw.get("temperature", 1.0))
            top_k = int(raw.get("top_k", 0))
            top_p = float(raw.get("top_p", 0.0))
            add_BOS = bool(raw.get("add_BOS", False))
            stop_on_double_eol = bool(raw.get("stop_on_double_eol", False))
            stop_on_eol = bool(raw.get("stop_on_eol", False))
            random_seed = int(raw.get("random_seed", -1))
            beam_width = raw.get("beam_width", None)
            stop_token = int(raw.get("stop_token", 0))
            length_penalty = float(raw.get("length_penalty", 1.0))

            with LOCK:
                try:
                    if beam_width is not None:
                        send_do_beam_search()
                        resp, segments, scores = beam_search_and_post_process(
                            model, prompts=prompts,
                            tokens_to_generate=tokens_to_generate,
                            beam_size=int(beam_width),
                            add_BOS=add_BOS, stop_token=stop_token,
                            num_return_gen=int(beam_width),
                            length_penalty=length_penalty,
                        )
                        return JSONResponse({
                            "text": resp, "segments": segments,
                            "scores": scores,
                        })
                    send_do_generate()
                    result = generate_and_post_process(
                        model, prompts=prompts,
                        tokens_to_generate=tokens_to_generate,
                        return_output_log_probs=logprobs,
                        top_k_sampling=top_k, top_p_sampling=top_p,
                        temperature=temperature, add_BOS=add_BOS,
                        stop_on_double_eol=stop_on_double_eol,
                        stop_on_eol=stop_on_eol, random_seed=random_seed,
                    )
                    resp, segments, output_log_probs, _ = result
                    return JSONResponse({
                        "text": resp, "segments": segments,
                        "logprobs": output_log_probs,
                    })
                except ValueError as e:
                    return JSONResponse({"message": str(e)}, status_code=400)

        return app

    def run(self, host="0.0.0.0", port=5000):
        import uvicorn

        app = self._build_app()
        uvicorn.run(app, host=host, port=port, log_level="warning")
