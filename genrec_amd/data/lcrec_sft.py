"""LCRec multi-task SFT sample generation.

Parity target: /root/reference/genrec/data/amazon_lcrec.py (690 LoC). Six
instruction-tuning tasks (amazon_lcrec.py:42-161, 214-221): seqrec,
item2index, index2item, fusionseqrec, itemsearch, preferenceobtain — each
with multiple prompt templates, per-task sampling weights, items rendered
as concatenated `<Ci_j>` codebook tokens, and the official numbered-history
format ("1. <...>, 2. <...>"). Eval samples are seqrec-only leave-one-out
(amazon_lcrec.py:432-454).

The synthetic variant generates item titles/descriptions procedurally so
the entire SFT pipeline runs offline.
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional, Sequence, Set

import numpy as np
from torch.utils.data import Dataset

from genrec_amd.config import ginlite
from genrec_amd.data.synthetic import _zipf_sequences

HISTORY_SEP = ", "

# Template inventory mirrors the reference's richness (~88 prompts across
# the six tasks, amazon_lcrec.py:42-161) with original wording; the
# item2index / index2item tasks carry title / description / combined
# variants like the reference's sub-groups.
TEMPLATES: Dict[str, List[str]] = {
    "seqrec": [  # 17 variants, parity with amazon_lcrec.py seqrec
        "User interaction history: {history}\nPredict the next item:",
        "The user has interacted with: {history}\nRecommend the next item:",
        "Given the ordered purchases {history}, which item comes next?",
        "Sequential behavior: {history}\nNext item:",
        "Considering the browsing sequence {history}, suggest the next item:",
        "These items were consumed in order: {history}\nThe next one will be:",
        "A shopper's timeline reads {history}. Predict their upcoming choice:",
        "Chronological activity: {history}\nForecast the next interaction:",
        "So far the user chose {history}. What follows?",
        "Observed sequence of items: {history}\nContinue the sequence with:",
        "From the log {history}, infer the item the user takes next:",
        "Recent engagements: {history}\nMost probable next item:",
        "Shopping trail: {history}\nComplete it with the next item:",
        "With a history of {history}, the recommendation engine should output:",
        "Interaction record: {history}\nNext-item prediction:",
        "The ordered basket history is {history}; name the follow-up item:",
        "Review the consumption order {history} and predict what is viewed next:",
    ],
    "item2index": [  # title -> index
        "Item title: {title}\nItem index:",
        "Which index identifies the product called \"{title}\"?",
        "Index lookup for: {title}\nResult:",
        "Translate the product name \"{title}\" into its index:",
        "The catalogue entry titled {title} is filed under index:",
        "Resolve \"{title}\" to an item index:",
    ],
    "item2index_desc": [  # description -> index
        "Item description: {description}\nItem index:",
        "A product described as \"{description}\" maps to index:",
        "From the blurb \"{description}\", recover the item index:",
        "Which index belongs to an item whose description reads {description}?",
        "Identify the index of the product matching: {description}",
        "Description to index: {description} ->",
    ],
    "item2index_combined": [  # title + description -> index
        "Map the item \"{title}\" ({description}) to its index:",
        "Product: {title}\nDetails: {description}\nIndex:",
        "Given the name \"{title}\" and the summary \"{description}\", the index is:",
        "Catalogue lookup — title: {title}; description: {description}; index:",
        "An item called {title}, characterised by {description}, is indexed as:",
        "Combine \"{title}\" with \"{description}\" and return the item index:",
        "For the listing {title} — {description}, output the index:",
    ],
    "index2item": [  # index -> title
        "Item index: {index}\nItem title:",
        "Which product does {index} refer to? Title:",
        "The index {index} belongs to the item titled:",
        "Decode {index} into a product name:",
        "Name the item stored under index {index}:",
        "Title of the catalogue entry {index}:",
    ],
    "index2item_desc": [  # index -> description
        "Item index: {index}\nItem description:",
        "Describe the item with index {index}:",
        "Summarise the product referenced by {index}:",
        "The entry {index} is described as:",
        "Provide the description recorded for index {index}:",
        "What does the item at {index} look like? Description:",
    ],
    "index2item_combined": [  # index -> title + description
        "Item index: {index}\nItem title and description:",
        "Expand {index} into the item's name and summary:",
        "Give both title and description for index {index}:",
        "The full catalogue record for {index} is:",
        "Report name plus details of the product indexed {index}:",
    ],
    "fusionseqrec": [  # 12 variants
        "History with titles: {history_with_titles}\nPredict the next item index:",
        "The user bought {history_with_titles}. Recommend the next item:",
        "Given purchases {history_with_titles}, the next item index is:",
        "Annotated history: {history_with_titles}\nNext item (index):",
        "Named interactions so far: {history_with_titles}\nForecast the next index:",
        "Shopping log with names: {history_with_titles}\nUpcoming item:",
        "From the titled sequence {history_with_titles}, predict the next identifier:",
        "Titled trail: {history_with_titles}\nContinue with the next index:",
        "Considering the labelled purchases {history_with_titles}, recommend next:",
        "Record (titles included): {history_with_titles}\nNext recommendation:",
        "The detailed history {history_with_titles} suggests the next item is:",
        "With item names {history_with_titles}, output the next sem-id:",
    ],
    "itemsearch": [  # 11 variants (query + optional history context)
        "A user wants: {query}. The best matching item index is:",
        "Search request: {query}\nRecommended item:",
        "Find an item for the preference \"{query}\":",
        "Query: {query}\nUser history: {history}\nBest match:",
        "Someone is shopping for {query}; given their past items {history}, return:",
        "Retrieve a product satisfying \"{query}\" for a user who chose {history}:",
        "Need: {query}\nContext: {history}\nItem:",
        "The request \"{query}\" combined with history {history} resolves to:",
        "Personalised search — intent: {query}; profile: {history}; answer:",
        "Match the wish \"{query}\" against the catalogue. Result index:",
        "Customer asks for {query}. Respond with the fitting item:",
    ],
    "preferenceobtain": [  # 12 variants
        "Based on the history {history}, summarize what the user prefers:",
        "Given interactions {history}, the user's preference can be described as:",
        "From the items {history}, characterise this user's taste:",
        "What does the sequence {history} reveal about the user's interests?",
        "Profile the shopper whose log is {history}:",
        "Derive a preference statement from {history}:",
        "The purchases {history} indicate a liking for:",
        "Interpret {history} as a description of user preferences:",
        "Looking at {history}, this user tends to enjoy:",
        "Distil the interests hidden in {history}:",
        "Given the engagement record {history}, their favourite kind of product is:",
        "User history {history}. Preference summary:",
    ],
}

DEFAULT_TASK_WEIGHTS = {
    "seqrec": 1.0, "item2index": 0.5, "index2item": 0.5,
    "fusionseqrec": 0.5, "itemsearch": 0.3, "preferenceobtain": 0.3,
}


def sem_ids_to_tokens(sem_ids: Sequence[int]) -> str:
    return "".join(f"<C{c}_{code}>" for c, code in enumerate(sem_ids))


class LCRecSFTDatasetBase(Dataset):
    """Generates {prompt, response, task} samples from item sequences +
    per-item sem-IDs + per-item text."""

    def __init__(self, sequences: List[List[int]],
                 item_sem_ids: np.ndarray,           # [N+1, C]
                 item_titles: List[str], item_descs: List[str],
                 split: str = "train", max_history: int = 20,
                 enabled_tasks: Optional[Set[str]] = None,
                 task_sample_weights: Optional[Dict[str, float]] = None,
                 add_prefix: bool = True, seed: int = 0,
                 max_samples: Optional[int] = None) -> None:
        self.item_sem_ids = item_sem_ids
        self.titles = item_titles
        self.descs = item_descs
        self.add_prefix = add_prefix
        self.n_codebooks = item_sem_ids.shape[1]
        rng = random.Random(seed)
        tasks = enabled_tasks or set(DEFAULT_TASK_WEIGHTS)
        weights = task_sample_weights or DEFAULT_TASK_WEIGHTS

        self.samples: List[Dict] = []
        if split == "train":
            for seq in sequences:
                hist_full = seq[:-2]
                if len(hist_full) < 2:
                    continue
                for i in range(1, len(hist_full)):
                    lo = max(0, i - max_history)
                    hist, target = hist_full[lo:i], hist_full[i]
                    for task in tasks:
                        if rng.random() > weights.get(task, 0.0):
                            continue
                        self.samples.append(
                            self._make(task, hist, target, rng))
        else:
            for seq in sequences:
                s = seq[:-1] if split == "valid" else seq
                if len(s) < 2:
                    continue
                lo = max(0, len(s) - 1 - max_history)
                self.samples.append(
                    self._make("seqrec", s[lo:-1], s[-1], rng))
        if max_samples is not None:
            self.samples = self.samples[:max_samples]

    # -------------------------------------------------- sample builders

    def _hist_tokens(self, hist: List[int]) -> str:
        toks = []
        for idx, it in enumerate(hist):
            t = sem_ids_to_tokens(self.item_sem_ids[it])
            toks.append(f"{idx + 1}. {t}" if self.add_prefix else t)
        return HISTORY_SEP.join(toks)

    @staticmethod
    def _pick_variant(task: str, rng: random.Random) -> str:
        """item2index/index2item spread over title/desc/combined subgroups
        (reference sub-group structure, amazon_lcrec.py:60-113), weighted
        by template count."""
        groups = [task, f"{task}_desc", f"{task}_combined"]
        weights = [len(TEMPLATES[g]) for g in groups]
        return rng.choices(groups, weights=weights)[0]

    def _make(self, task: str, hist: List[int], target: int,
              rng: random.Random) -> Dict:
        tgt_tokens = sem_ids_to_tokens(self.item_sem_ids[target])
        if task in ("item2index", "index2item"):
            group = self._pick_variant(task, rng)
        else:
            group = task
        tmpl = rng.choice(TEMPLATES[group])
        if task == "seqrec":
            prompt = tmpl.format(history=self._hist_tokens(hist))
            response = tgt_tokens
        elif task == "item2index":
            prompt = tmpl.format(title=self.titles[target],
                                 description=self.descs[target])
            response = tgt_tokens
        elif task == "index2item":
            prompt = tmpl.format(index=tgt_tokens)
            if group.endswith("_desc"):
                response = self.descs[target]
            elif group.endswith("_combined"):
                response = f"{self.titles[target]}. {self.descs[target]}"
            else:
                response = self.titles[target]
        elif task == "fusionseqrec":
            hwt = HISTORY_SEP.join(
                f"{i + 1}. {self.titles[it]} ({sem_ids_to_tokens(self.item_sem_ids[it])})"
                for i, it in enumerate(hist))
            prompt = tmpl.format(history_with_titles=hwt)
            response = tgt_tokens
        elif task == "itemsearch":
            prompt = tmpl.format(query=self.descs[target],
                                 history=self._hist_tokens(hist))
            response = tgt_tokens
        elif task == "preferenceobtain":
            prompt = tmpl.format(history=self._hist_tokens(hist))
            response = self.descs[target]
        else:
            raise ValueError(task)
        return {"prompt": prompt, "response": response, "task": task,
                "target_sem_ids": [int(c) for c in self.item_sem_ids[target]],
                "target_title": self.titles[target]}

    def __len__(self) -> int:
        return len(self.samples)

    def __getitem__(self, idx: int) -> Dict:
        return self.samples[idx]


@ginlite.configurable(name="SyntheticLCRecDataset")
class SyntheticLCRecDataset(LCRecSFTDatasetBase):
    def __init__(self, num_users: int = 500, num_items: int = 2000,
                 mean_len: float = 8.9, sem_id_dim: int = 5,
                 codebook_size: int = 256, split: str = "train",
                 seed: int = 0, max_history: int = 20,
                 max_samples: Optional[int] = None, **kw) -> None:
        rng = np.random.default_rng(seed + 33)
        sem = rng.integers(0, codebook_size, size=(num_items + 1, sem_id_dim))
        adjectives = ["Deluxe", "Mini", "Organic", "Smart", "Classic",
                      "Portable", "Premium", "Eco"]
        nouns = ["Moisturizer", "Shampoo", "Lotion", "Serum", "Cream",
                 "Cleanser", "Balm", "Oil"]
        titles = ["<pad>"] + [
            f"{adjectives[i % len(adjectives)]} {nouns[(i // 7) % len(nouns)]} #{i}"
            for i in range(1, num_items + 1)]
        descs = ["<pad>"] + [
            f"a {adjectives[(i + 3) % len(adjectives)].lower()} product for daily use, id {i}"
            for i in range(1, num_items + 1)]
        seqs = _zipf_sequences(num_users, num_items, mean_len, seed)
        super().__init__(seqs, sem, titles, descs, split=split,
                         max_history=max_history, seed=seed,
                         max_samples=max_samples, **kw)
        self.sem_id_dim = sem_id_dim
        self.codebook_size = codebook_size
