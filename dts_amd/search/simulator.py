"""Multi-turn conversation rollout.

Parity: reference backend/core/dts/components/simulator.py:55-474 —
expand_nodes with the linear shortcut (ref :123-125), intent
scatter-gather forking (ref :136-186), per-task timeout (ref :199-214),
sequential user→terminate-check→assistant turns (ref :234-305), rephrased
first message with turn-0 user-sim skip (ref :316-354), empty-response
retry (ref :414-447) and the termination heuristics (ref :34-52, 449-460).

MI355X note: each expansion task is the natural unit of DP branch sharding
(SURVEY.md §2.4); the per-turn calls all land in one continuous-batching
scheduler, so sibling branches' shared conversation prefixes hit the same
paged KV blocks (serving/kv_cache.py).
"""

from __future__ import annotations

import asyncio
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.errors import EmptyResponseError, LLMError
from dts_amd.llm.types import Message
from dts_amd.search import prompts
from dts_amd.search.config import GenerationBudget
from dts_amd.search.events import create_event_emitter
from dts_amd.search.tree import DialogueTree, derive_node_id, generate_node_id
from dts_amd.search.types import DialogueNode, NodeStatus, Strategy, UserIntent
from dts_amd.utils.logging import log_phase, logger

TERMINATION_SIGNALS = [
    "goodbye",
    "bye",
    "i'm done",
    "i have to go",
    "thanks, bye",
    "i'm leaving",
    "end conversation",
    "stop",
    "quit",
    "exit",
    "i give up",
    "forget it",
    "never mind",
    "this isn't working",
    "i'm confused",
    "you're not helping",
    "i don't understand",
]

#: per-expansion-task wall budget (ref simulator.py:199: 120 s/task)
TASK_TIMEOUT_S = 120.0


class ConversationSimulator:
    def __init__(
        self,
        llm: LLM,
        goal: str,
        model: Optional[str] = None,
        temperature: float = 0.7,
        max_concurrency: int = 16,
        on_usage: Optional[Callable[[Any, str], None]] = None,
        on_event: Optional[Callable[[str, dict], Any]] = None,
        budget: Optional[GenerationBudget] = None,
        seed: Optional[int] = None,
        reasoning_enabled: bool = False,
    ) -> None:
        self.llm = llm
        self.goal = goal
        self.model = model
        self.reasoning_enabled = reasoning_enabled
        self.temperature = temperature
        self.budget = budget or GenerationBudget()
        self.seed = seed
        self._sem = asyncio.Semaphore(max_concurrency)
        self._on_usage = on_usage
        self._emit = create_event_emitter(on_event)

    # ------------------------------------------------------------------
    async def expand_nodes(
        self,
        nodes: list,
        turns: int,
        intents_per_node: int = 1,
        tree: Optional[DialogueTree] = None,
        generate_intents: Optional[Callable] = None,
    ) -> list:
        if intents_per_node <= 1 or generate_intents is None:
            return await self._expand_linear_batch(nodes, turns)

        log_phase(
            "FORK",
            f"Generating {intents_per_node} intents for {len(nodes)} nodes...",
            indent=1,
        )
        intent_results = await asyncio.gather(
            *[generate_intents(node.messages, intents_per_node) for node in nodes],
            return_exceptions=True,
        )

        expansion_tasks = []
        fallback_nodes = []
        creation_order: dict = {}
        for node, intents in zip(nodes, intent_results):
            if isinstance(intents, Exception) or not intents:
                logger.warning("Intent generation failed for %s; linear expansion", node.id)
                fallback_nodes.append(node)
                continue
            strategy_name = node.strategy.tagline if node.strategy else "root"
            for idx, intent in enumerate(intents):
                self._emit(
                    "intent_generated",
                    {
                        "strategy": strategy_name,
                        "index": idx + 1,
                        "total": len(intents),
                        "label": intent.label,
                        "emotional_tone": intent.emotional_tone,
                        "cognitive_stance": intent.cognitive_stance,
                    },
                )
                child = DialogueNode(
                    id=derive_node_id(node.id, f"intent:{idx}"),
                    parent_id=node.id,
                    depth=node.depth + 1,
                    strategy=node.strategy,
                    user_intent=intent,
                    messages=list(node.messages),
                )
                if tree is not None:
                    tree.add_child(node.id, child)
                expansion_tasks.append(self._expand_with_intent(child, turns, intent))
                creation_order[child.id] = len(creation_order)

        for node in fallback_nodes:
            expansion_tasks.append(self._expand_linear(node, turns))
            creation_order[node.id] = len(creation_order)

        log_phase("FORK", f"Expanding {len(expansion_tasks)} branches...", indent=1)
        total_timeout = TASK_TIMEOUT_S * max(1, len(expansion_tasks))

        expanded: list = []
        failed = 0
        try:
            for coro in asyncio.as_completed(expansion_tasks, timeout=total_timeout):
                try:
                    result = await coro
                    if isinstance(result, DialogueNode):
                        expanded.append(result)
                except (TimeoutError, asyncio.TimeoutError):
                    logger.warning("Expansion timed out")
                    failed += 1
                except Exception as e:  # noqa: BLE001
                    logger.error("Expansion error: %s", e)
                    failed += 1
        except (TimeoutError, asyncio.TimeoutError):
            logger.warning("Expansion batch timed out")
        log_phase("FORK", f"Completed: {len(expanded)} | Failed: {failed}", indent=1)
        # as_completed yields in COMPLETION order; downstream the sibling
        # order reaches the comparative-judge prompt, so a latency wiggle
        # would change judge inputs run to run. Restore creation order —
        # the reference has the same incidental nondeterminism
        # (ref simulator.py:199-214); determinism is a deliberate
        # improvement here (seeded runs must reproduce).
        expanded.sort(key=lambda n: creation_order.get(n.id, 1 << 30))
        return expanded

    async def _expand_linear_batch(self, nodes: list, turns: int) -> list:
        results = await asyncio.gather(
            *[self._expand_linear(node, turns) for node in nodes],
            return_exceptions=True,
        )
        expanded = []
        for node, result in zip(nodes, results):
            if isinstance(result, Exception):
                logger.error("Error expanding %s: %s", node.id, result)
                node.status = NodeStatus.ERROR
            else:
                expanded.append(result)
        return expanded

    # ------------------------------------------------------------------
    async def _run_turn(
        self,
        node: DialogueNode,
        history: list,
        turn_idx: int,
        skip_user_simulation: bool = False,
        label: Optional[str] = None,
    ) -> bool:
        """One user+assistant exchange; returns False when expansion stops."""
        if not skip_user_simulation:
            try:
                user_response = await self._simulate_user(history)
            except EmptyResponseError:
                node.status = NodeStatus.ERROR
                node.prune_reason = "empty user response after retries"
                return False
            history.append(Message.user(user_response))
            if self._should_terminate(user_response):
                node.status = NodeStatus.TERMINAL
                return False

        try:
            assistant_response = await self._generate_assistant(history, node.strategy)
        except EmptyResponseError:
            node.status = NodeStatus.ERROR
            node.prune_reason = "empty assistant response after retries"
            return False
        history.append(Message.assistant(assistant_response))
        return True

    async def _expand_linear(self, node: DialogueNode, turns: int) -> DialogueNode:
        history = list(node.messages)
        for turn_idx in range(turns):
            if not await self._run_turn(node, history, turn_idx):
                break
        node.messages = history
        return node

    async def _expand_with_intent(
        self, node: DialogueNode, turns: int, first_intent: UserIntent
    ) -> DialogueNode:
        history = list(node.messages)
        if history and history[0].role == "user":
            try:
                rephrased = await self._rephrase_initial_message(
                    history[0].content or "", first_intent
                )
                history[0] = Message.user(rephrased)
            except EmptyResponseError:
                pass  # keep original on rephrase failure (ref simulator.py:338-345)
        for turn_idx in range(turns):
            skip_user = turn_idx == 0
            if not await self._run_turn(
                node, history, turn_idx, skip_user, first_intent.label
            ):
                break
        node.messages = history
        return node

    # ------------------------------------------------------------------
    async def _rephrase_initial_message(
        self, original: str, intent: UserIntent
    ) -> str:
        system, user = prompts.rephrase_with_intent(
            original_message=original,
            intent_label=intent.label,
            intent_description=intent.description,
            emotional_tone=intent.emotional_tone,
            cognitive_stance=intent.cognitive_stance,
        )
        return await self._call_with_retry(
            [Message.system(system), Message.user(user)],
            phase="rephrase",
            max_tokens=self.budget.rephrase,
        )

    async def _simulate_user(
        self, history: list, intent: Optional[UserIntent] = None
    ) -> str:
        intent_dict = None
        if intent:
            intent_dict = {
                "label": intent.label,
                "description": intent.description,
                "emotional_tone": intent.emotional_tone,
                "cognitive_stance": intent.cognitive_stance,
            }
        system, user = prompts.user_simulation(
            conversation_goal=self.goal, user_intent=intent_dict
        )
        messages = [Message.system(system)] + history + [Message.user(user)]
        return await self._call_with_retry(
            messages, phase="user", max_tokens=self.budget.user
        )

    async def _generate_assistant(
        self, history: list, strategy: Optional[Strategy]
    ) -> str:
        system, user = prompts.assistant_continuation(
            conversation_goal=self.goal,
            strategy_tagline=strategy.tagline if strategy else "",
            strategy_description=strategy.description if strategy else "",
        )
        max_tokens = self.budget.assistant
        if self.reasoning_enabled:
            # local semantics for the wire flag the reference drops
            # (SURVEY.md §4.1.1): the actor may think in <think> spans,
            # which LLM.complete strips from the returned content; the
            # budget grows to cover the hidden reasoning
            system += (
                "\n\nBefore answering, reason privately inside "
                "<think>...</think> tags; the user sees only what follows "
                "the closing tag."
            )
            max_tokens *= 2
        messages = [Message.system(system)] + history + [Message.user(user)]
        return await self._call_with_retry(
            messages, phase="assistant", max_tokens=max_tokens
        )

    async def _call_with_retry(
        self,
        messages: list,
        phase: str,
        max_tokens: int,
        max_retries: int = 3,
    ) -> str:
        """Retry empty responses with backoff (ref simulator.py:414-447)."""
        delay = 0.5
        for attempt in range(1, max_retries + 1):
            try:
                completion = await self._call_llm(messages, phase, max_tokens, attempt)
            except LLMError as e:
                if attempt == max_retries:
                    raise EmptyResponseError(str(e)) from e
                await asyncio.sleep(min(delay, 4.0))
                delay *= 2
                continue
            content = completion.message.content
            if content and content.strip():
                return content
            if attempt < max_retries:
                await asyncio.sleep(min(delay, 4.0))
                delay *= 2
        raise EmptyResponseError(f"Empty response for phase '{phase}'")

    def _should_terminate(self, user_response: str) -> bool:
        response_lower = user_response.lower().strip()
        if any(signal in response_lower for signal in TERMINATION_SIGNALS):
            return True
        return len(response_lower) < 20 and any(
            w in response_lower for w in ["no", "nope", "wrong", "bad", "ugh"]
        )

    async def _call_llm(
        self, messages: list, phase: str, max_tokens: int, attempt: int
    ):
        async with self._sem:
            seed = None
            if self.seed is not None:
                # distinct, run-stable seed per call; the last message's
                # content decorrelates sibling branches at the same turn
                from dts_amd.utils.seeding import stable_seed

                seed = stable_seed(
                    self.seed,
                    phase,
                    len(messages),
                    attempt,
                    messages[-1].content if messages else "",
                )
            completion = await self.llm.complete(
                messages,
                model=self.model,
                temperature=self.temperature,
                max_tokens=max_tokens,
                seed=seed,
            )
            if self._on_usage:
                self._on_usage(completion, phase)
            return completion
