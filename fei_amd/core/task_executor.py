"""TaskExecutor: the autonomous multi-iteration loop.

Parity: reference fei/core/task_executor.py — repeats ``assistant.chat``
until the ``[TASK_COMPLETE]`` sentinel appears or max_iterations is hit
(task_executor.py:205-260), recovering tool output when the model answers
empty (task_executor.py:111-155). The reference slept 0.5 s per iteration
(task_executor.py:252) purely to pace a remote API; a local engine needs no
pacing, so the default delay is 0.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from fei_amd.utils.logging import get_logger

logger = get_logger("core.task_executor")

COMPLETION_SIGNAL = "[TASK_COMPLETE]"

TASK_INSTRUCTION = (
    "You are executing a multi-step task. Work step by step using tools. "
    f"When the task is fully complete, include the exact text {COMPLETION_SIGNAL} "
    "in your reply."
)


@dataclass
class TaskContext:
    task: str
    max_iterations: int = 10
    iterations_done: int = 0
    complete: bool = False
    responses: List[str] = field(default_factory=list)
    started_at: float = field(default_factory=time.time)
    elapsed_s: float = 0.0


class TaskExecutor:
    def __init__(self, assistant, iteration_delay_s: float = 0.0):
        self.assistant = assistant
        self.iteration_delay_s = iteration_delay_s

    def _process_response(self, ctx: TaskContext, response: Optional[str]) -> str:
        """Detect/strip the completion sentinel; recover tool output on empty
        responses (reference: task_executor.py:67-155)."""
        if not response:
            response = self.assistant.conversation.scrape_tool_output()
        if response and COMPLETION_SIGNAL in response:
            ctx.complete = True
            response = response.replace(COMPLETION_SIGNAL, "").strip()
        return response or ""

    def execute_task(
        self,
        task: str,
        max_iterations: int = 10,
        system_prompt: Optional[str] = None,
    ) -> Dict[str, Any]:
        """Run the loop (reference: task_executor.py:205-260)."""
        ctx = TaskContext(task=task, max_iterations=max_iterations)
        prompt = f"{TASK_INSTRUCTION}\n\nTask: {task}"
        current = prompt
        while ctx.iterations_done < ctx.max_iterations and not ctx.complete:
            t0 = time.perf_counter()
            response = self.assistant.chat(current, system_prompt=system_prompt)
            response = self._process_response(ctx, response)
            ctx.responses.append(response)
            ctx.iterations_done += 1
            logger.debug("task iteration %d took %.3fs", ctx.iterations_done,
                         time.perf_counter() - t0)
            current = "Continue with the next step of the task."
            if self.iteration_delay_s and not ctx.complete:
                time.sleep(self.iteration_delay_s)
        ctx.elapsed_s = time.time() - ctx.started_at
        return {
            "task": task,
            "complete": ctx.complete,
            "iterations": ctx.iterations_done,
            "responses": ctx.responses,
            "final_response": ctx.responses[-1] if ctx.responses else "",
            "elapsed_s": ctx.elapsed_s,
        }

    def execute_interactive(self, task: str, on_response=None,
                            max_iterations: int = 10,
                            system_prompt: Optional[str] = None) -> Dict[str, Any]:
        """Like execute_task, invoking ``on_response(iteration, text)`` after
        each step (reference: task_executor.py:262-317)."""
        ctx = TaskContext(task=task, max_iterations=max_iterations)
        current = f"{TASK_INSTRUCTION}\n\nTask: {task}"
        while ctx.iterations_done < ctx.max_iterations and not ctx.complete:
            response = self._process_response(
                ctx, self.assistant.chat(current, system_prompt=system_prompt))
            ctx.responses.append(response)
            ctx.iterations_done += 1
            if on_response is not None:
                on_response(ctx.iterations_done, response)
            current = "Continue with the next step of the task."
        ctx.elapsed_s = time.time() - ctx.started_at
        return {
            "task": task, "complete": ctx.complete,
            "iterations": ctx.iterations_done, "responses": ctx.responses,
            "final_response": ctx.responses[-1] if ctx.responses else "",
            "elapsed_s": ctx.elapsed_s,
        }
