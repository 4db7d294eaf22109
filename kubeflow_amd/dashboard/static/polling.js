// polling.js — ExponentialBackoff poller, behavior parity with the
// reference's UI-driven reconcile-by-polling (kubeflow-common-lib
// exponential-backoff.ts:16-70 + index-default.component.ts:50-61):
// poll at `interval` for `retries` ticks, then double up to `maxInterval`;
// reset() returns to fast polling — callers reset whenever the fetched
// data CHANGED, so active resources keep a snappy UI while idle lists
// back off to one request per 16 s.
export class ExponentialBackoff {
  constructor(fn, { interval = 1000, retries = 3, maxInterval = 16000 } = {}) {
    this.fn = fn;
    this.base = interval;
    this.retries = retries;
    this.maxInterval = maxInterval;
    this.stopped = true;
    this._timer = null;
    this._n = 0;
  }

  currentInterval() {
    const doublings = Math.max(0, Math.floor(this._n / this.retries));
    return Math.min(this.base * 2 ** doublings, this.maxInterval);
  }

  start() {
    this.stopped = false;
    this._n = 0;
    this._tick();
    return this;
  }

  async _tick() {
    if (this.stopped) return;
    try {
      await this.fn();
    } catch {
      /* fn handles its own errors; keep polling */
    }
    if (this.stopped) return;
    this._n += 1;
    this._timer = setTimeout(() => this._tick(), this.currentInterval());
  }

  reset() {
    this._n = 0;
  }

  stop() {
    this.stopped = true;
    if (this._timer) clearTimeout(this._timer);
  }
}

// helper: JSON change detection for reset-on-change semantics
export function changed(prev, next) {
  return JSON.stringify(prev) !== JSON.stringify(next);
}
