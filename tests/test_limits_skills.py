"""Direct tests for chat/limits.py (TPMRateLimiter + retry ladder) and
tools/skills.py (SKILL.md frontmatter + skills.json discovery)."""

import json
import os

from senweaver_amd.chat.limits import (
    CHAT_RETRIES,
    TPMRateLimiter,
    get_retry_delay_ms,
    is_context_length_error,
    is_rate_limit_error,
)
from senweaver_amd.tools.skills import SkillService, parse_frontmatter


def test_retry_ladder_constants():
    assert CHAT_RETRIES == 5
    # TPM (429) errors: exponential from 3 s, capped at 60 s
    assert get_retry_delay_ms(0, True) == 3000
    assert get_retry_delay_ms(1, True) == 6000
    assert get_retry_delay_ms(10, True) == 60000
    # other errors: gentler 1.5^ ladder, capped at 30 s
    assert get_retry_delay_ms(1, False) == 3000
    assert get_retry_delay_ms(2, False) == 4500
    assert get_retry_delay_ms(20, False) == 30000


def test_error_classifiers():
    assert is_context_length_error("Error: maximum context length is 8192")
    assert is_context_length_error("400 Bad Request: too many tokens")
    assert not is_context_length_error("500 internal error")
    assert is_rate_limit_error("429 Too Many Requests")
    assert is_rate_limit_error("rate_limit_exceeded")
    assert not is_rate_limit_error("connection reset")


def test_rate_limiter_reactive_cooldown():
    t = [0.0]
    rl = TPMRateLimiter(clock=lambda: t[0])
    assert rl.get_wait_time_ms("default") == 0
    # a 429 with retry-after wins over the backoff ladder
    cd = rl.handle_rate_limit_error("default", '429 {"retry_after": 2.5}')
    assert cd == 2500
    assert rl.get_wait_time_ms("default") == 2500
    t[0] = 1000
    assert rl.get_wait_time_ms("default") == 1500
    # success clears the cooldown (reactive, not predictive)
    rl.record_success("default")
    assert rl.get_wait_time_ms("default") == 0
    # without retry-after: the TPM ladder value
    cd2 = rl.handle_rate_limit_error("default", "429 slow down", attempt=1)
    assert cd2 == 6000


def test_rate_limiter_min_interval():
    t = [0.0]
    rl = TPMRateLimiter(clock=lambda: t[0])
    rl.record_request("default")  # default config: minRequestInterval 100 ms
    assert rl.get_wait_time_ms("default") == 100
    t[0] = 40
    assert rl.get_wait_time_ms("default") == 60
    t[0] = 200
    assert rl.get_wait_time_ms("default") == 0
    # the local backbone has no interval
    rl.record_request("local")
    assert rl.get_wait_time_ms("local") == 0


def test_frontmatter_parse():
    meta, body = parse_frontmatter(
        "---\nname: refactor\ndescription: 'How to refactor'\n---\nBody here\n")
    assert meta == {"name": "refactor", "description": "How to refactor"}
    assert body == "Body here\n"
    meta2, body2 = parse_frontmatter("no frontmatter at all")
    assert meta2 == {} and body2 == "no frontmatter at all"


def test_skill_discovery(tmp_path):
    base = tmp_path / ".senweaver" / "skills"
    (base / "deploy").mkdir(parents=True)
    (base / "deploy" / "SKILL.md").write_text(
        "---\nname: deploy\ndescription: Ship it\n---\nRun the pipeline.\n")
    (base / "review.md").write_text("---\nname: review\n---\nCheck the diff.\n")
    (base / "skills.json").write_text(json.dumps([
        {"name": "triage", "description": "Bug triage", "content": "Steps."},
        {"name": "deploy", "description": "dup ignored", "content": "x"},
    ]))
    svc = SkillService(str(tmp_path))
    names = sorted(s.name for s in svc.list_skills())
    assert names == ["deploy", "review", "triage"]
    assert svc.get_skill("deploy").content.strip() == "Run the pipeline."
    assert svc.get_skill("deploy").description == "Ship it"  # SKILL.md wins
    assert svc.get_skill("triage").description == "Bug triage"
    assert svc.get_skill("missing") is None


def test_skill_discovery_empty(tmp_path):
    svc = SkillService(str(tmp_path))
    assert svc.list_skills() == []
