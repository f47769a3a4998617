"""Petri-net bot: queries fan into answers, two answers trigger a summary.
Each transition firing runs as its own container."""

import beta9_amd


def answer(markers=None, **kw):
    q = markers["queries"][0]["q"]
    return {"markers": {"answers": [{"text": f"answer to: {q}"}]}}


def summarize(markers=None, **kw):
    texts = [m["text"] for m in markers["answers"]]
    return {"markers": {"summaries": [{"text": " | ".join(texts)}]}}


bot = beta9_amd.Bot(name="research", welcome_message="ask me two things")
bot.location("queries", marker={"q": "str"})
bot.location("answers", marker={"text": "str"})
bot.location("summaries", marker={"text": "str"})
bot.transition(inputs={"queries": 1}, outputs=["answers"])(answer)
bot.transition(inputs={"answers": 2}, outputs=["summaries"])(summarize)

if __name__ == "__main__":
    session = bot.create_session()
    session.push("queries", {"q": "why is the sky blue"})
    session.push("queries", {"q": "why is water wet"})
    for ev in session.events(timeout=60):
        print(ev["type"], {k: v for k, v in ev.items() if k not in ("type", "ts")})
        if ev["type"] == "transition_complete" and ev["transition"] == "summarize":
            break
    print(session.state()["markers"]["summaries"])
