"""OpenAI-style serving API demo (in-process TestClient; run `fei api`
for a real uvicorn server)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fastapi.testclient import TestClient
from fei_amd.serve.api import create_app

client = TestClient(create_app(model="llama3-tiny"))
print(client.get("/health").json())

r = client.post("/v1/completions",
                json={"prompt": "def main():", "max_tokens": 16})
print("completion:", r.json()["choices"][0]["text"][:60])

with client.stream("POST", "/v1/completions",
                   json={"prompt": "stream this", "max_tokens": 12,
                         "stream": True, "stop_on_eos": False}) as s:
    n = sum(1 for line in s.iter_lines() if line.startswith("data: "))
print(f"streamed {n} SSE events")

sid = client.post("/v1/sessions",
                  json={"prompt": "session demo", "max_tokens": 8}).json()["session_id"]
while client.post("/v1/sessions/step?n=4").json()["active"]:
    pass
print("session result:", client.get(f"/v1/sessions/{sid}").json()["done"])
