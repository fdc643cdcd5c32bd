"""Simple CLI client for the text-generation server (reference
tools/text_generation_cli.py)."""

import json
import sys
import urllib.request


if __name__ == "__main__":
    url = f"http://{sys.argv[1]}/api"
    while True:
        sentence = input("Enter prompt: ")
        tokens_to_generate = int(input("Enter number of tokens to generate: "))
        data = json.dumps(
            {"prompts": [sentence], "tokens_to_generate": tokens_to_generate}
        ).encode("utf-8")
        req = urllib.request.Request(
            url, data=data, method="PUT",
            headers={"Content-Type": "application/json"},
        )
        with urllib.request.urlopen(req) as resp:
            body = json.loads(resp.read().decode("utf-8"))
            print(body["text"][0])
