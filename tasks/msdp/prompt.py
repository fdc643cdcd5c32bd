"""Multi-stage dialogue prompting (reference tasks/msdp/prompt.py):
prompt a pretrained GPT to generate the KNOWLEDGE for a dialogue turn
(stage 1) and then the RESPONSE grounded on it (stage 2).

Input samples are TSV lines `topic \t turn1 [SEP] turn2 ... [\t knowledge]`.
Knowledge prompts are a jsonl dict {"<topic> <last_turn>": [examples...]};
response prompts are a flat text file of examples. Generation can run on a
local checkpoint or against a running REST server (--megatron_api_url,
reference :19-35) — the wire format matches our inference server."""

from __future__ import annotations

import json

import torch

from megatron_amd import parallel as mpu
from megatron_amd.checkpointing import load_checkpoint
from megatron_amd.config import get_config
from megatron_amd.inference.api import generate_and_post_process
from megatron_amd.models import MODEL_CLASSES, ModelType
from megatron_amd.training import get_model
from megatron_amd.utils import print_rank_0


def call_model_api(inputs, tokens_to_generate):
    """PUT /api of a running text-generation server (same contract as
    megatron_amd.inference.server)."""
    import requests

    cfg = get_config()
    data = {"prompts": [inputs], "tokens_to_generate": tokens_to_generate,
            "top_k": 1}
    out = requests.put(
        cfg.megatron_api_url,
        headers={"Content-Type": "application/json; charset=UTF-8"},
        data=json.dumps(data),
    ).json()["text"][0]
    out = out[len(inputs):]
    return out.split("\n")[0].strip()


def read_prompts(prompt_path, prompt_type, n_example):
    if prompt_type == "knowledge":
        prompt_examples_dict = {}
        with open(prompt_path) as f:
            for line in f:
                line_dict = json.loads(line.strip())
                key = list(line_dict.keys())[0]
                if key not in prompt_examples_dict:
                    prompt_examples_dict[key] = "".join(
                        inst.strip() + " \n" for inst in line_dict[key]
                    )
        return prompt_examples_dict
    with open(prompt_path) as f:
        examples = f.readlines()[:n_example]
    return "".join(inst.strip() + " \n" for inst in examples)


def build_input(sample_line, prompt_type, knowledge_prompts, response_prompt):
    """Assemble the full prompt string for one test sample."""
    splits = sample_line.strip().split("\t")
    topic = splits[0]
    turns = splits[1].split(" [SEP] ")
    last_turn = turns[-1].strip()
    if prompt_type == "knowledge":
        key = topic + " " + last_turn
        inputs = knowledge_prompts[key]
        inputs += "( " + last_turn + " ) " + topic + " =>"
        return inputs
    knowledge = splits[2].strip()
    inputs = response_prompt
    inputs += "Topic: " + topic + ". "
    inputs += "User says: " + last_turn + " "
    inputs += "We know that: " + knowledge + " "
    inputs += "System replies:"
    return inputs


def model_provider(pre_process=True, post_process=True):
    cfg = get_config()
    print_rank_0("building GPT model ...")
    model_cls = MODEL_CLASSES[cfg.model_name or "gpt"]
    return model_cls(cfg, parallel_output=False, pre_process=pre_process,
                     post_process=post_process)


def generate_samples_by_prompting_input_from_file(model):
    cfg = get_config()
    assert cfg.sample_input_file is not None
    assert cfg.prompt_type in ("knowledge", "response")

    knowledge_prompts = response_prompt = None
    if cfg.prompt_type == "knowledge":
        knowledge_prompts = read_prompts(cfg.prompt_file, "knowledge",
                                         cfg.num_prompt_examples)
    else:
        response_prompt = read_prompts(cfg.prompt_file, "response",
                                       cfg.num_prompt_examples)

    with open(cfg.sample_input_file) as f:
        samples = f.readlines()
    out_path = cfg.sample_output_file or cfg.sample_input_file + ".out"

    model.eval()
    first = (
        mpu.is_pipeline_first_stage()
        and mpu.get_tensor_model_parallel_rank() == 0
    )
    fout = open(out_path, "w") if first else None
    with torch.no_grad():
        for line in samples:
            inputs = build_input(line, cfg.prompt_type, knowledge_prompts,
                                 response_prompt)
            prompts_plus, _, _, _ = generate_and_post_process(
                model, prompts=[inputs],
                tokens_to_generate=cfg.out_seq_length, top_k_sampling=1,
            )
            generation = prompts_plus[0][len(inputs):].split("\n")[0].strip()
            if fout:
                fout.write(generation + "\n")
    if fout:
        fout.close()
    print_rank_0(f"wrote generations to {out_path}")


def generate_samples_by_calling_api():
    cfg = get_config()
    knowledge_prompts = response_prompt = None
    if cfg.prompt_type == "knowledge":
        knowledge_prompts = read_prompts(cfg.prompt_file, "knowledge",
                                         cfg.num_prompt_examples)
    else:
        response_prompt = read_prompts(cfg.prompt_file, "response",
                                       cfg.num_prompt_examples)
    with open(cfg.sample_input_file) as f, \
            open(cfg.sample_output_file, "w") as fout:
        for line in f:
            inputs = build_input(line, cfg.prompt_type, knowledge_prompts,
                                 response_prompt)
            fout.write(call_model_api(inputs, cfg.out_seq_length) + "\n")


def main():
    cfg = get_config()
    if getattr(cfg, "api_prompt", False):
        generate_samples_by_calling_api()
        return
    model = get_model(model_provider, ModelType.encoder_or_decoder,
                      wrap_with_ddp=False, cfg=cfg)
    if cfg.load is not None:
        load_checkpoint(model, None, None, cfg)
    assert len(model) == 1
    generate_samples_by_prompting_input_from_file(model[0])
