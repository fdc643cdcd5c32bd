"""MSDP dataset preprocessing (reference tasks/msdp/preprocessing.py,
condensed to the Wizard-of-Wikipedia pipeline): convert the raw WoW json
into the `topic \\t dialogue-context \\t knowledge \\t response` TSV the
prompting stages consume, plus the knowledge/response reference files used
by MSDP-EVAL-F1.

  python tasks/msdp/preprocessing.py --func process_wow_dataset \\
      --raw_file data.json --processed_file out.tsv \\
      [--knwl_ref_file k.txt --resp_ref_file r.txt]

(nltk's word_tokenize is replaced by whitespace tokenization — nltk is not
in this image; F1 normalization lowercases and strips punctuation anyway.)
"""

from __future__ import annotations

import argparse
import json


def _simple_tokenize(text):
    return " ".join(text.split())


def process_wow_dataset(raw_file, processed_file, knwl_ref_file=None,
                        resp_ref_file=None):
    """WoW json -> TSV; wizard turns produce one sample each, with the
    checked sentence as golden knowledge and the checked passage (or the
    chosen topic) as the topic."""
    print(f"> Loading data from {raw_file}")
    with open(raw_file) as fr:
        dialog_data = json.load(fr)

    fproc = open(processed_file, "w")
    fknwl = open(knwl_ref_file, "w") if knwl_ref_file else None
    fresp = open(resp_ref_file, "w") if resp_ref_file else None

    for sample in dialog_data:
        turn_list = []
        for j, turn in enumerate(sample["dialog"]):
            text = turn["text"]
            if not text.endswith(("?", ".", "!")):
                text += "."
            if j == 0:
                turn_list.append(text)
                continue

            speaker = turn["speaker"].lower()
            if "wizard" in speaker:
                checked_sentence = list(turn["checked_sentence"].values())
                checked_passage = list(turn["checked_passage"].values())
                assert len(checked_sentence) <= 1
                knowledge = (checked_sentence[0] if checked_sentence
                             else "no_passages_used")
                topic = (checked_passage[0] if len(checked_passage) == 1
                         else sample["chosen_topic"])
                if topic == "no_passages_used":
                    topic = sample["chosen_topic"]

                dialog_context = " [SEP] ".join(turn_list)
                response = text
                turn_list.append(response)

                fproc.write(f"{topic}\t{dialog_context}\t{knowledge}\t"
                            f"{response}\n")
                if fknwl:
                    fknwl.write(knowledge + "\n")
                if fresp:
                    fresp.write(_simple_tokenize(response) + "\n")
            else:
                assert "apprentice" in speaker
                turn_list.append(text)

    fproc.close()
    if fknwl:
        fknwl.close()
    if fresp:
        fresp.close()


def build_knowledge_prompts(processed_file, prompt_file, n_examples=10):
    """Group the processed TSV by (topic, last turn) and emit the jsonl
    knowledge-generation prompt dictionary (reference get_database /
    prompt-construction, condensed): each key maps to up to n_examples
    '( last_turn ) topic => knowledge' instances drawn from other samples
    of the same topic."""
    by_topic = {}
    entries = []
    with open(processed_file) as f:
        for line in f:
            topic, context, knowledge, _ = line.rstrip("\n").split("\t")
            last_turn = context.split(" [SEP] ")[-1]
            inst = f"( {last_turn} ) {topic} => {knowledge}"
            by_topic.setdefault(topic, []).append(inst)
            entries.append((topic, last_turn))

    with open(prompt_file, "w") as f:
        for topic, last_turn in entries:
            key = f"{topic} {last_turn}"
            examples = by_topic[topic][:n_examples]
            f.write(json.dumps({key: examples}) + "\n")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--func", required=True,
                   choices=["process_wow_dataset", "build_knowledge_prompts"])
    p.add_argument("--raw_file")
    p.add_argument("--processed_file", required=True)
    p.add_argument("--knwl_ref_file", default=None)
    p.add_argument("--resp_ref_file", default=None)
    p.add_argument("--prompt_file", default=None)
    p.add_argument("--num_prompt_examples", type=int, default=10)
    args = p.parse_args()
    if args.func == "process_wow_dataset":
        process_wow_dataset(args.raw_file, args.processed_file,
                            args.knwl_ref_file, args.resp_ref_file)
    else:
        build_knowledge_prompts(args.processed_file, args.prompt_file,
                                args.num_prompt_examples)


if __name__ == "__main__":
    main()
