"""Upload a converted HF-format checkpoint directory to the HuggingFace Hub
(reference tools/push_to_hub.py). Requires network + credentials at runtime."""

import argparse


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("input_dir", help="HF-format checkpoint directory "
                        "(output of weights_conversion/megatron_to_hf.py)")
    parser.add_argument("--hf_repo_name", required=True)
    parser.add_argument("--auth_token", default=None)
    args = parser.parse_args()

    from huggingface_hub import HfApi

    api = HfApi(token=args.auth_token)
    api.create_repo(args.hf_repo_name, exist_ok=True)
    api.upload_folder(
        folder_path=args.input_dir, repo_id=args.hf_repo_name,
        repo_type="model",
    )
    print(f"pushed {args.input_dir} to {args.hf_repo_name}")


if __name__ == "__main__":
    main()
