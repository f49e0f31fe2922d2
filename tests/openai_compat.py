"""OpenAI SDK compatibility check against a running dnet_amd API
(reference: tests/openai_compat.py). Run manually:

    python tests/openai_compat.py [base_url]

Skipped as a pytest module unless the `openai` package is installed and a
server is reachable.
"""
import sys

import pytest

openai = pytest.importorskip("openai")


def run(base_url="http://localhost:8080/v1"):
    client = openai.OpenAI(base_url=base_url, api_key="dnet")
    models = client.models.list()
    assert models.data, "no models listed"
    resp = client.chat.completions.create(
        model="tiny-random", max_tokens=8,
        messages=[{"role": "user", "content": "hello"}])
    assert resp.choices[0].message is not None
    stream = client.chat.completions.create(
        model="tiny-random", max_tokens=8, stream=True,
        messages=[{"role": "user", "content": "hello"}])
    chunks = list(stream)
    assert chunks, "no stream chunks"
    print("openai compat ok:", len(chunks), "chunks")


if __name__ == "__main__":
    run(*(sys.argv[1:] or []))
