"""MetricName mirror — the label-set operations the binary-operator and
aggregation matching needs (lib/storage/metric_name.go:32 MetricName
{MetricGroup []byte; Tags []Tag}).

Only the semantics the vmselect hot path uses are mirrored:
RemoveTagsOn/RemoveTagsIgnoring (metric_name.go:247,281), RemoveTag (:265),
SetTags/setAllTags (:317,362), SetTagBytes (:352), ResetMetricGroup (:240),
GetTagValue (:299).  `marshal_sorted()` is the grouping key
(marshalMetricNameSorted, app/vmselect/promql/timeseries.go:281): two names
compare equal iff their metric group and sorted tag sets are equal — the
exact byte layout is internal to this engine (the reference's is too; only
equality semantics matter for map keys).
"""


class MetricName:
    __slots__ = ("metric_group", "tags")

    def __init__(self, metric_group=b"", tags=None):
        if isinstance(metric_group, str):
            metric_group = metric_group.encode()
        self.metric_group = bytes(metric_group)
        # tags: list of (key bytes, value bytes); insertion order preserved
        self.tags = []
        for k, v in (tags or []):
            self.add_tag(k, v)

    @staticmethod
    def _b(x):
        return x.encode() if isinstance(x, str) else bytes(x)

    def copy(self):
        m = MetricName()
        m.metric_group = self.metric_group
        m.tags = list(self.tags)
        return m

    def reset_metric_group(self):
        self.metric_group = b""

    def add_tag(self, key, value):
        self.tags.append((self._b(key), self._b(value)))

    def set_tag(self, key, value):
        # SetTagBytes (metric_name.go:352): replace in place or append
        key, value = self._b(key), self._b(value)
        for i, (k, _) in enumerate(self.tags):
            if k == key:
                self.tags[i] = (key, value)
                return
        self.tags.append((key, value))

    def get_tag_value(self, key):
        key = self._b(key)
        if key == b"__name__":
            return self.metric_group
        for k, v in self.tags:
            if k == key:
                return v
        return None

    def remove_tag(self, key):
        key = self._b(key)
        if key == b"__name__":
            self.reset_metric_group()
            return
        self.tags = [(k, v) for k, v in self.tags if k != key]

    def remove_tags_on(self, on_tags):
        # RemoveTagsOn (metric_name.go:247): keep only tags in on_tags;
        # MetricGroup survives only when __name__ is listed.
        on = {self._b(t) for t in on_tags}
        if b"__name__" not in on:
            self.reset_metric_group()
        if not on:
            self.tags = []
            return
        self.tags = [(k, v) for k, v in self.tags if k in on]

    def remove_tags_ignoring(self, ignoring_tags):
        # RemoveTagsIgnoring (metric_name.go:281)
        ig = {self._b(t) for t in ignoring_tags}
        if not ig:
            return
        if b"__name__" in ig:
            self.reset_metric_group()
        self.tags = [(k, v) for k, v in self.tags if k not in ig]

    def set_tags(self, add_tags, prefix, skip_tags, src):
        # SetTags (metric_name.go:317) — group_left(...) prefix copying
        prefix = self._b(prefix)
        skip = {self._b(t) for t in skip_tags}
        if len(add_tags) == 1 and self._b(add_tags[0]) == b"*":
            for k, v in src.tags:
                if k in skip:
                    continue
                self.set_tag(prefix + k, v)
            return
        for name in add_tags:
            name = self._b(name)
            if name in skip:
                continue
            if name == b"__name__":
                self.metric_group = src.metric_group
                continue
            sv = None
            for k, v in src.tags:
                if k == name:
                    sv = v
                    break
            if sv is None:
                self.remove_tag(name)
                continue
            self.set_tag(prefix + name, sv)

    def marshal_sorted(self):
        parts = [self.metric_group]
        for k, v in sorted(self.tags):
            parts.append(k)
            parts.append(v)
        return b"\x00".join(parts)

    def __repr__(self):
        tags = ",".join(f"{k.decode()}={v.decode()!r}" for k, v in self.tags)
        return f"{self.metric_group.decode()}{{{tags}}}"

    def __eq__(self, other):
        return (isinstance(other, MetricName) and
                self.marshal_sorted() == other.marshal_sorted())

    def __hash__(self):
        return hash(self.marshal_sorted())
