"""Bot petri-net orchestration (reference analog:
pkg/abstractions/experimental/bot/)."""

import asyncio

import pytest

import beta9_amd
from tests.helpers import cluster


@pytest.mark.asyncio
async def test_bot_transition_fires_and_cascades():
    async with cluster(n_workers=1) as c:
        bot = beta9_amd.Bot(name="research", welcome_message="hi")
        bot.location("queries", marker={"q": "str"})
        bot.location("answers", marker={"text": "str"})
        bot.location("summaries", marker={"text": "str"})

        import handlers

        bot.transition(inputs={"queries": 1}, outputs=["answers"])(handlers.bot_answer)
        bot.transition(inputs={"answers": 2}, outputs=["summaries"])(handlers.bot_summarize)

        session = await asyncio.to_thread(bot.create_session)
        await asyncio.to_thread(session.push, "queries", {"q": "one"})
        await asyncio.to_thread(session.push, "queries", {"q": "two"})

        seen = []
        def drain():
            for ev in session.events(timeout=20.0):
                seen.append(ev)
                if ev.get("type") == "transition_complete" and ev.get("transition") == "bot_summarize":
                    return
        await asyncio.to_thread(drain)

        types = [e["type"] for e in seen]
        assert "welcome" in types
        assert types.count("transition_fired") >= 3  # 2x answer + 1x summarize (cascade)
        st = await asyncio.to_thread(session.state)
        assert len(st["markers"]["summaries"]) == 1
        assert st["markers"]["queries"] == [] and st["markers"]["answers"] == []


@pytest.mark.asyncio
async def test_bot_confirm_gate():
    async with cluster(n_workers=1) as c:
        import handlers

        bot = beta9_amd.Bot(name="gated")
        bot.location("inbox")
        bot.location("outbox")
        bot.transition(inputs={"inbox": 1}, outputs=["outbox"], confirm=True)(handlers.bot_answer)
        session = await asyncio.to_thread(bot.create_session)
        await asyncio.to_thread(session.push, "inbox", {"q": "x"})
        await asyncio.sleep(0.3)
        st = await asyncio.to_thread(session.state)
        assert len(st["markers"]["inbox"]) == 1  # held: awaiting confirm
        await asyncio.to_thread(session.confirm, "bot_answer")

        def drain():
            for ev in session.events(timeout=20.0):
                if ev.get("type") == "transition_complete":
                    return ev
        ev = await asyncio.to_thread(drain)
        assert ev is not None
