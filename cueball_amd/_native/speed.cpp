/*
 * cueball_amd._speed: native event/FSM runtime core.
 *
 * The framework's hot loop (a pool claim/release cycle) walks six-plus
 * Moore-machine state transitions across three FSMs, each with scoped
 * listener registration/teardown and async stateChanged queueing (see
 * cueball_amd/fsm.py for the semantics contract, which this module
 * reproduces exactly -- the whole pytest suite runs against either
 * implementation).  This C++ core removes the interpreter overhead from
 * EventEmitter dispatch, scope bookkeeping and the transition loop.
 *
 * The reference (node-cueball) has no native code; this is the one
 * component where a native core is justified by measurement (see
 * profiles/README.md): event-loop throughput is the library's
 * performance story.
 *
 * Plain C++/CPython API -- no GPU code exists in this problem domain
 * (SURVEY.md section 0).
 */

#define PY_SSIZE_T_CLEAN
#include <Python.h>
#include <structmember.h>
#include <cmath>

namespace {

/* interned strings, created at module init */
PyObject *s_stateChanged;
PyObject *s_listener;
PyObject *s_on;
PyObject *s_remove_listener;
PyObject *s_call_soon;
PyObject *s_call_later;
PyObject *s_cancel;
PyObject *s_state_prefix;      /* "state_" */
PyObject *s_dot;               /* "." */
PyObject *s_underscore;        /* "_" */
PyObject *s_flush_name;        /* "_flush_state_changed" */
PyObject *s_internal;          /* "_cueball_internal" */
PyObject *s_is_closed;         /* "is_closed" */

PyObject *g_get_loop;          /* python helper: get_loop(loop) */
PyObject *g_fsm_error;         /* exception class FSMError */
PyObject *g_entry_name_cache;  /* dict: state name -> "state_x_y" */
PyObject *g_flush_batches;     /* dict: loop -> _FlushBatch */
PyObject *g_tracer;            /* optional fn(fsm, state) on transitions */
PyObject *g_remove_desc;       /* EmitterType's remove_listener descriptor */

/* ------------------------------------------------------------------ */
/* EventEmitter                                                        */
/* ------------------------------------------------------------------ */

typedef struct {
    PyObject_HEAD
    PyObject *ev_events;   /* dict: str -> list of callables */
    PyObject *ev_dict;     /* instance __dict__ (lazy) */
    PyObject *ev_weakrefs;
} Emitter;

extern PyTypeObject EmitterType;

int
Emitter_init(PyObject *self_, PyObject *args, PyObject *kwds)
{
    Emitter *self = (Emitter *)self_;
    (void)args; (void)kwds;
    if (self->ev_events == NULL) {
        self->ev_events = PyDict_New();
        if (self->ev_events == NULL)
            return -1;
    }
    return 0;
}

int
Emitter_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Emitter *self = (Emitter *)self_;
    Py_VISIT(self->ev_events);
    Py_VISIT(self->ev_dict);
    return 0;
}

int
Emitter_clear_(PyObject *self_)
{
    Emitter *self = (Emitter *)self_;
    Py_CLEAR(self->ev_events);
    Py_CLEAR(self->ev_dict);
    return 0;
}

void
Emitter_dealloc(PyObject *self_)
{
    Emitter *self = (Emitter *)self_;
    PyTypeObject *tp = Py_TYPE(self_);
    PyObject_GC_UnTrack(self_);
    if (self->ev_weakrefs != NULL)
        PyObject_ClearWeakRefs(self_);
    Emitter_clear_(self_);
    tp->tp_free(self_);
}

/* core: register a listener */
PyObject *
emitter_add(Emitter *self, PyObject *event, PyObject *listener)
{
    if (self->ev_events == NULL) {
        self->ev_events = PyDict_New();
        if (self->ev_events == NULL)
            return NULL;
    }
    PyObject *ls = PyDict_GetItemWithError(self->ev_events, event);
    if (ls == NULL) {
        if (PyErr_Occurred())
            return NULL;
        ls = PyList_New(0);
        if (ls == NULL)
            return NULL;
        if (PyDict_SetItem(self->ev_events, event, ls) < 0) {
            Py_DECREF(ls);
            return NULL;
        }
        Py_DECREF(ls);  /* dict holds it */
    }
    if (PyList_Append(ls, listener) < 0)
        return NULL;
    Py_INCREF(listener);
    return listener;
}

PyObject *
Emitter_on(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "on(event, listener)");
        return NULL;
    }
    return emitter_add((Emitter *)self_, args[0], args[1]);
}

/* once() wrapper object */
typedef struct {
    PyObject_HEAD
    PyObject *ow_emitter;   /* weak semantics not needed; strong ref */
    PyObject *ow_event;
    PyObject *ow_listener;
} OnceWrapper;

extern PyTypeObject OnceWrapperType;

PyObject *
OnceWrapper_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    OnceWrapper *self = (OnceWrapper *)self_;
    (void)kwds;
    /* remove ourselves first (reference events.py once semantics) */
    PyObject *res = PyObject_CallMethodObjArgs(
        self->ow_emitter, s_remove_listener, self->ow_event,
        (PyObject *)self, NULL);
    if (res == NULL)
        return NULL;
    Py_DECREF(res);
    return PyObject_Call(self->ow_listener, args, NULL);
}

PyObject *
OnceWrapper_get_listener(PyObject *self_, void *closure)
{
    OnceWrapper *self = (OnceWrapper *)self_;
    (void)closure;
    Py_INCREF(self->ow_listener);
    return self->ow_listener;
}

int
OnceWrapper_traverse(PyObject *self_, visitproc visit, void *arg)
{
    OnceWrapper *self = (OnceWrapper *)self_;
    Py_VISIT(self->ow_emitter);
    Py_VISIT(self->ow_event);
    Py_VISIT(self->ow_listener);
    return 0;
}

int
OnceWrapper_clear_(PyObject *self_)
{
    OnceWrapper *self = (OnceWrapper *)self_;
    Py_CLEAR(self->ow_emitter);
    Py_CLEAR(self->ow_event);
    Py_CLEAR(self->ow_listener);
    return 0;
}

void
OnceWrapper_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    OnceWrapper_clear_(self_);
    PyObject_GC_Del(self_);
}

PyGetSetDef OnceWrapper_getset[] = {
    {(char *)"listener", OnceWrapper_get_listener, NULL,
     (char *)"original listener", NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject OnceWrapperType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._OnceWrapper",       /* tp_name */
    sizeof(OnceWrapper),                      /* tp_basicsize */
    0,                                        /* tp_itemsize */
    OnceWrapper_dealloc,                      /* tp_dealloc */
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    OnceWrapper_call,                         /* tp_call */
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,  /* tp_flags */
    0,                                        /* tp_doc */
    OnceWrapper_traverse,                     /* tp_traverse */
    OnceWrapper_clear_,                       /* tp_clear */
    0, 0, 0, 0, 0, 0,
    OnceWrapper_getset,                       /* tp_getset */
};

PyObject *
Emitter_once(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "once(event, listener)");
        return NULL;
    }
    OnceWrapper *w = PyObject_GC_New(OnceWrapper, &OnceWrapperType);
    if (w == NULL)
        return NULL;
    Py_INCREF(self_);
    w->ow_emitter = self_;
    Py_INCREF(args[0]);
    w->ow_event = args[0];
    Py_INCREF(args[1]);
    w->ow_listener = args[1];
    PyObject_GC_Track((PyObject *)w);
    PyObject *r = emitter_add((Emitter *)self_, args[0], (PyObject *)w);
    if (r == NULL) {
        Py_DECREF(w);
        return NULL;
    }
    Py_DECREF(r);
    return (PyObject *)w;
}

PyObject *
Emitter_remove_listener(PyObject *self_, PyObject *const *args,
                        Py_ssize_t nargs)
{
    Emitter *self = (Emitter *)self_;
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "remove_listener(event, listener)");
        return NULL;
    }
    if (self->ev_events == NULL)
        Py_RETURN_NONE;
    PyObject *ls = PyDict_GetItemWithError(self->ev_events, args[0]);
    if (ls == NULL) {
        if (PyErr_Occurred())
            return NULL;
        Py_RETURN_NONE;
    }
    Py_ssize_t n = PyList_GET_SIZE(ls);
    Py_ssize_t found = -1;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *item = PyList_GET_ITEM(ls, i);
        if (item == args[1]) {
            found = i;
            break;
        }
        /* bound methods are fresh objects per attribute access: fall
         * back to == (matches list.remove semantics in the Python
         * implementation) */
        int eq = PyObject_RichCompareBool(item, args[1], Py_EQ);
        if (eq < 0)
            return NULL;
        if (eq) {
            found = i;
            break;
        }
    }
    if (found < 0) {
        /* allow removing a once() registration by its inner listener */
        for (Py_ssize_t i = 0; i < n; i++) {
            PyObject *w = PyList_GET_ITEM(ls, i);
            if (Py_TYPE(w) != &OnceWrapperType)
                continue;
            PyObject *inner = ((OnceWrapper *)w)->ow_listener;
            if (inner == args[1]) {
                found = i;
                break;
            }
            int eq = PyObject_RichCompareBool(inner, args[1], Py_EQ);
            if (eq < 0)
                return NULL;
            if (eq) {
                found = i;
                break;
            }
        }
    }
    if (found >= 0) {
        if (PyList_SetSlice(ls, found, found + 1, NULL) < 0)
            return NULL;
        if (PyList_GET_SIZE(ls) == 0) {
            if (PyDict_DelItem(self->ev_events, args[0]) < 0)
                PyErr_Clear();
        }
    }
    Py_RETURN_NONE;
}

PyObject *
Emitter_remove_all_listeners(PyObject *self_, PyObject *const *args,
                             Py_ssize_t nargs)
{
    Emitter *self = (Emitter *)self_;
    if (self->ev_events == NULL)
        Py_RETURN_NONE;
    if (nargs == 0 || args[0] == Py_None) {
        PyDict_Clear(self->ev_events);
    } else {
        if (PyDict_DelItem(self->ev_events, args[0]) < 0)
            PyErr_Clear();
    }
    Py_RETURN_NONE;
}

PyObject *
Emitter_listeners(PyObject *self_, PyObject *event)
{
    Emitter *self = (Emitter *)self_;
    if (self->ev_events != NULL) {
        PyObject *ls = PyDict_GetItemWithError(self->ev_events, event);
        if (ls != NULL)
            return PyList_GetSlice(ls, 0, PyList_GET_SIZE(ls));
        if (PyErr_Occurred())
            return NULL;
    }
    return PyList_New(0);
}

PyObject *
Emitter_listener_count(PyObject *self_, PyObject *event)
{
    Emitter *self = (Emitter *)self_;
    Py_ssize_t n = 0;
    if (self->ev_events != NULL) {
        PyObject *ls = PyDict_GetItemWithError(self->ev_events, event);
        if (ls != NULL)
            n = PyList_GET_SIZE(ls);
        else if (PyErr_Occurred())
            return NULL;
    }
    return PyLong_FromSsize_t(n);
}

PyObject *
Emitter_event_names(PyObject *self_, PyObject *noargs)
{
    Emitter *self = (Emitter *)self_;
    (void)noargs;
    if (self->ev_events == NULL)
        return PyList_New(0);
    return PyDict_Keys(self->ev_events);
}

/* emit with node snapshot semantics */
int
emitter_emit_core(Emitter *self, PyObject *event, PyObject *const *eargs,
                  Py_ssize_t neargs)
{
    if (self->ev_events == NULL)
        return 0;
    PyObject *ls = PyDict_GetItemWithError(self->ev_events, event);
    if (ls == NULL)
        return PyErr_Occurred() ? -1 : 0;
    Py_ssize_t n = PyList_GET_SIZE(ls);
    if (n == 0)
        return 0;
    if (n == 1) {
        /* copy-free fast path (snapshot still holds: a listener added
         * during the call is not invoked) */
        PyObject *cb = PyList_GET_ITEM(ls, 0);
        Py_INCREF(cb);
        PyObject *r = PyObject_Vectorcall(cb, eargs, neargs, NULL);
        Py_DECREF(cb);
        if (r == NULL)
            return -1;
        Py_DECREF(r);
        return 1;
    }
    PyObject *snap = PyList_GetSlice(ls, 0, n);
    if (snap == NULL)
        return -1;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *cb = PyList_GET_ITEM(snap, i);
        PyObject *r = PyObject_Vectorcall(cb, eargs, neargs, NULL);
        if (r == NULL) {
            Py_DECREF(snap);
            return -1;
        }
        Py_DECREF(r);
    }
    Py_DECREF(snap);
    return 1;
}

PyObject *
Emitter_emit(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs < 1) {
        PyErr_SetString(PyExc_TypeError, "emit(event, *args)");
        return NULL;
    }
    int r = emitter_emit_core((Emitter *)self_, args[0], args + 1,
                              nargs - 1);
    if (r < 0)
        return NULL;
    return PyBool_FromLong(r);
}

PyMethodDef Emitter_methods[] = {
    {"on", (PyCFunction)(void (*)(void))Emitter_on, METH_FASTCALL, NULL},
    {"add_listener", (PyCFunction)(void (*)(void))Emitter_on,
     METH_FASTCALL, NULL},
    {"once", (PyCFunction)(void (*)(void))Emitter_once, METH_FASTCALL,
     NULL},
    {"remove_listener",
     (PyCFunction)(void (*)(void))Emitter_remove_listener, METH_FASTCALL,
     NULL},
    {"remove_all_listeners",
     (PyCFunction)(void (*)(void))Emitter_remove_all_listeners,
     METH_FASTCALL, NULL},
    {"listeners", Emitter_listeners, METH_O, NULL},
    {"listener_count", Emitter_listener_count, METH_O, NULL},
    {"event_names", Emitter_event_names, METH_NOARGS, NULL},
    {"emit", (PyCFunction)(void (*)(void))Emitter_emit, METH_FASTCALL,
     NULL},
    {NULL, NULL, 0, NULL},
};

PyMemberDef Emitter_members[] = {
    {(char *)"_events", T_OBJECT, offsetof(Emitter, ev_events), READONLY,
     NULL},
    {NULL, 0, 0, 0, NULL},
};

PyTypeObject EmitterType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.EventEmitter",        /* tp_name */
    sizeof(Emitter),                          /* tp_basicsize */
    0,                                        /* tp_itemsize */
    Emitter_dealloc,                          /* tp_dealloc */
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_BASETYPE | Py_TPFLAGS_HAVE_GC,
    "node-style EventEmitter (native)",       /* tp_doc */
    Emitter_traverse,                         /* tp_traverse */
    Emitter_clear_,                           /* tp_clear */
    0,                                        /* tp_richcompare */
    offsetof(Emitter, ev_weakrefs),           /* tp_weaklistoffset */
    0, 0,
    Emitter_methods,                          /* tp_methods */
    Emitter_members,                          /* tp_members */
    0, 0, 0, 0, 0,
    offsetof(Emitter, ev_dict),               /* tp_dictoffset */
    Emitter_init,                             /* tp_init */
    0,
    PyType_GenericNew,                        /* tp_new */
};

/* ------------------------------------------------------------------ */
/* forward decls                                                       */
/* ------------------------------------------------------------------ */

typedef struct FSMOb FSMOb;
int fsm_goto_state(FSMOb *fsm, PyObject *state);

/* ------------------------------------------------------------------ */
/* StateScope                                                          */
/* ------------------------------------------------------------------ */

typedef struct {
    PyObject_HEAD
    PyObject *sc_fsm;       /* FSMOb*, strong */
    PyObject *sc_listeners; /* flat list [em, evt, cb, ...] or NULL */
    PyObject *sc_timers;    /* list of cancellables or NULL */
    int sc_active;
} Scope;

extern PyTypeObject ScopeType;

/* guarded zero/N-arg callback: no-op once scope inactive */
typedef struct {
    PyObject_HEAD
    PyObject *gc_scope;   /* Scope* */
    PyObject *gc_cb;
} GuardedCb;

extern PyTypeObject GuardedCbType;

PyObject *
GuardedCb_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    GuardedCb *self = (GuardedCb *)self_;
    (void)kwds;
    if (!((Scope *)self->gc_scope)->sc_active)
        Py_RETURN_NONE;
    return PyObject_Call(self->gc_cb, args, NULL);
}

int
GuardedCb_traverse(PyObject *self_, visitproc visit, void *arg)
{
    GuardedCb *self = (GuardedCb *)self_;
    Py_VISIT(self->gc_scope);
    Py_VISIT(self->gc_cb);
    return 0;
}

int
GuardedCb_clear_(PyObject *self_)
{
    GuardedCb *self = (GuardedCb *)self_;
    Py_CLEAR(self->gc_scope);
    Py_CLEAR(self->gc_cb);
    return 0;
}

void
GuardedCb_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    GuardedCb_clear_(self_);
    PyObject_GC_Del(self_);
}

PyTypeObject GuardedCbType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._GuardedCb",
    sizeof(GuardedCb),
    0,
    GuardedCb_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    GuardedCb_call,
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    0,
    GuardedCb_traverse,
    GuardedCb_clear_,
};

PyObject *
make_guarded(PyObject *scope, PyObject *cb)
{
    GuardedCb *g = PyObject_GC_New(GuardedCb, &GuardedCbType);
    if (g == NULL)
        return NULL;
    Py_INCREF(scope);
    g->gc_scope = scope;
    Py_INCREF(cb);
    g->gc_cb = cb;
    PyObject_GC_Track((PyObject *)g);
    return (PyObject *)g;
}

int
scope_check_active(Scope *self, const char *what)
{
    if (!self->sc_active) {
        PyErr_Format(g_fsm_error, "%s used on exited state scope", what);
        return -1;
    }
    return 0;
}

/* the registration core of S.on(), callable from C (SlotKit) */
int
scope_on_c(Scope *self, PyObject *emitter, PyObject *event, PyObject *cb)
{
    PyObject *r;
    if (Py_TYPE(emitter) == &EmitterType) {
        r = emitter_add((Emitter *)emitter, event, cb);
    } else {
        r = PyObject_CallMethodObjArgs(emitter, s_on, event, cb, NULL);
    }
    if (r == NULL)
        return -1;
    Py_DECREF(r);
    if (self->sc_listeners == NULL) {
        self->sc_listeners = PyList_New(0);
        if (self->sc_listeners == NULL)
            return -1;
    }
    if (PyList_Append(self->sc_listeners, emitter) < 0 ||
        PyList_Append(self->sc_listeners, event) < 0 ||
        PyList_Append(self->sc_listeners, cb) < 0)
        return -1;
    return 0;
}

PyObject *
Scope_on(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    Scope *self = (Scope *)self_;
    if (nargs != 3) {
        PyErr_SetString(PyExc_TypeError, "S.on(emitter, event, cb)");
        return NULL;
    }
    if (scope_check_active(self, "S.on()") < 0)
        return NULL;
    if (scope_on_c(self, args[0], args[1], args[2]) < 0)
        return NULL;
    Py_RETURN_NONE;
}

int
scope_add_timer(Scope *self, PyObject *handle)
{
    if (self->sc_timers == NULL) {
        self->sc_timers = PyList_New(0);
        if (self->sc_timers == NULL)
            return -1;
    }
    return PyList_Append(self->sc_timers, handle);
}

PyObject *fsm_get_loop_of(FSMOb *fsm);

PyObject *
Scope_timeout(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    Scope *self = (Scope *)self_;
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "S.timeout(ms, cb)");
        return NULL;
    }
    if (scope_check_active(self, "S.timeout()") < 0)
        return NULL;
    double ms = PyFloat_AsDouble(args[0]);
    if (ms == -1.0 && PyErr_Occurred())
        return NULL;
    PyObject *guarded = make_guarded(self_, args[1]);
    if (guarded == NULL)
        return NULL;
    PyObject *secs = PyFloat_FromDouble(ms / 1000.0);
    if (secs == NULL) {
        Py_DECREF(guarded);
        return NULL;
    }
    PyObject *loop = fsm_get_loop_of((FSMOb *)self->sc_fsm);
    PyObject *handle = PyObject_CallMethodObjArgs(loop, s_call_later, secs,
                                                  guarded, NULL);
    Py_DECREF(secs);
    Py_DECREF(guarded);
    if (handle == NULL)
        return NULL;
    int rc = scope_add_timer(self, handle);
    Py_DECREF(handle);
    if (rc < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
Scope_immediate(PyObject *self_, PyObject *cb)
{
    Scope *self = (Scope *)self_;
    if (scope_check_active(self, "S.immediate()") < 0)
        return NULL;
    PyObject *guarded = make_guarded(self_, cb);
    if (guarded == NULL)
        return NULL;
    PyObject *loop = fsm_get_loop_of((FSMOb *)self->sc_fsm);
    PyObject *handle = PyObject_CallMethodObjArgs(loop, s_call_soon,
                                                  guarded, NULL);
    Py_DECREF(guarded);
    if (handle == NULL)
        return NULL;
    int rc = scope_add_timer(self, handle);
    Py_DECREF(handle);
    if (rc < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
Scope_callback(PyObject *self_, PyObject *cb)
{
    Scope *self = (Scope *)self_;
    if (scope_check_active(self, "S.callback()") < 0)
        return NULL;
    return make_guarded(self_, cb);
}

PyObject *Scope_interval(PyObject *self_, PyObject *const *args,
                         Py_ssize_t nargs);

PyObject *
Scope_valid_transitions(PyObject *self_, PyObject *states);

PyObject *
Scope_goto_state(PyObject *self_, PyObject *state)
{
    Scope *self = (Scope *)self_;
    if (!self->sc_active)
        Py_RETURN_NONE;   /* stale handler during a cascade: obsolete */
    if (fsm_goto_state((FSMOb *)self->sc_fsm, state) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
Scope_dispose(PyObject *self_, PyObject *noargs)
{
    Scope *self = (Scope *)self_;
    (void)noargs;
    self->sc_active = 0;
    PyObject *ls = self->sc_listeners;
    self->sc_listeners = NULL;
    if (ls != NULL) {
        Py_ssize_t n = PyList_GET_SIZE(ls);
        for (Py_ssize_t i = 0; i + 2 < n; i += 3) {
            PyObject *em = PyList_GET_ITEM(ls, i);
            PyObject *evt = PyList_GET_ITEM(ls, i + 1);
            PyObject *cb = PyList_GET_ITEM(ls, i + 2);
            PyObject *r;
            /* fast path only when the type's MRO resolves
             * remove_listener to our own C method (no override) */
            int fast = 0;
            if (PyType_IsSubtype(Py_TYPE(em), &EmitterType)) {
                PyObject *desc = _PyType_Lookup(Py_TYPE(em),
                                                s_remove_listener);
                fast = (desc == g_remove_desc);
            }
            if (fast) {
                PyObject *cargs[2] = {evt, cb};
                r = Emitter_remove_listener(em, cargs, 2);
            } else {
                r = PyObject_CallMethodObjArgs(em, s_remove_listener, evt,
                                               cb, NULL);
            }
            if (r == NULL) {
                Py_DECREF(ls);
                return NULL;
            }
            Py_DECREF(r);
        }
        Py_DECREF(ls);
    }
    PyObject *ts = self->sc_timers;
    self->sc_timers = NULL;
    if (ts != NULL) {
        Py_ssize_t n = PyList_GET_SIZE(ts);
        for (Py_ssize_t i = 0; i < n; i++) {
            PyObject *r = PyObject_CallMethodObjArgs(
                PyList_GET_ITEM(ts, i), s_cancel, NULL);
            if (r == NULL) {
                Py_DECREF(ts);
                return NULL;
            }
            Py_DECREF(r);
        }
        Py_DECREF(ts);
    }
    Py_RETURN_NONE;
}

PyObject *
Scope_get_active(PyObject *self_, void *closure)
{
    (void)closure;
    return PyBool_FromLong(((Scope *)self_)->sc_active);
}

int
Scope_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Scope *self = (Scope *)self_;
    Py_VISIT(self->sc_fsm);
    Py_VISIT(self->sc_listeners);
    Py_VISIT(self->sc_timers);
    return 0;
}

int
Scope_clear_(PyObject *self_)
{
    Scope *self = (Scope *)self_;
    Py_CLEAR(self->sc_fsm);
    Py_CLEAR(self->sc_listeners);
    Py_CLEAR(self->sc_timers);
    return 0;
}

void
Scope_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    Scope_clear_(self_);
    PyObject_GC_Del(self_);
}

PyMethodDef Scope_methods[] = {
    {"on", (PyCFunction)(void (*)(void))Scope_on, METH_FASTCALL, NULL},
    {"timeout", (PyCFunction)(void (*)(void))Scope_timeout, METH_FASTCALL,
     NULL},
    {"interval", (PyCFunction)(void (*)(void))Scope_interval,
     METH_FASTCALL, NULL},
    {"immediate", Scope_immediate, METH_O, NULL},
    {"callback", Scope_callback, METH_O, NULL},
    {"valid_transitions", Scope_valid_transitions, METH_O, NULL},
    {"goto_state", Scope_goto_state, METH_O, NULL},
    {"_dispose", Scope_dispose, METH_NOARGS, NULL},
    {NULL, NULL, 0, NULL},
};

PyGetSetDef Scope_getset[] = {
    {(char *)"active", Scope_get_active, NULL, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject ScopeType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.StateScope",
    sizeof(Scope),
    0,
    Scope_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    "FSM state scope (native)",
    Scope_traverse,
    Scope_clear_,
    0, 0, 0, 0,
    Scope_methods,
    0,
    Scope_getset,
};

/* ------------------------------------------------------------------ */
/* FSM                                                                 */
/* ------------------------------------------------------------------ */

struct FSMOb {
    Emitter base;
    PyObject *f_loop;
    PyObject *f_state;          /* str or NULL */
    PyObject *f_scope;          /* Scope* or NULL */
    PyObject *f_valid;          /* sequence or NULL */
    PyObject *f_pending;        /* str or NULL */
    PyObject *f_emit_queue;     /* list */
    PyObject *f_history;        /* list */
    PyObject *f_flush_bound;    /* cached bound _flush_state_changed */
    PyObject *f_fastkit;        /* SlotKit* or NULL: native entries for
                                 * the busy/idle hot cycle */
    int f_entering;
    int f_emit_scheduled;
};

/* forward decl (SlotKit lives after the CH/Queue sections);
 * returns 1 = state entered natively, 0 = use the python entry,
 * -1 = error */
int slotkit_maybe_entry(PyObject *kit_, FSMOb *fsm, PyObject *target,
                        Scope *scope);

/* claim-handle terminal cleanup (defined near the CH section) */
extern PyObject *t_empty;
extern PyTypeObject CHType;
extern PyTypeObject CTType;
void ch_terminal_cleanup(FSMOb *self);

extern PyTypeObject FSMType;

PyObject *
fsm_get_loop_of(FSMOb *fsm)
{
    return fsm->f_loop;
}

int fsm_schedule_flush(FSMOb *self);

/* Per-FSM drain cap: one FSM whose listeners keep triggering its own
 * transitions would otherwise drain forever inside a single loop
 * callback (the batch cap only bounds distinct flush entries).  The
 * remainder is rescheduled in order. */
#define FSM_FLUSH_CAP 64

int
fsm_flush_core(FSMOb *self)
{
    self->f_emit_scheduled = 0;
    int n = 0;
    while (PyList_GET_SIZE(self->f_emit_queue) > 0 &&
           n++ < FSM_FLUSH_CAP) {
        PyObject *st = PyList_GET_ITEM(self->f_emit_queue, 0);
        Py_INCREF(st);
        if (PyList_SetSlice(self->f_emit_queue, 0, 1, NULL) < 0) {
            Py_DECREF(st);
            return -1;
        }
        PyObject *eargs[2] = {s_stateChanged, st};
        int r = emitter_emit_core(&self->base, eargs[0], eargs + 1, 1);
        Py_DECREF(st);
        if (r < 0)
            return -1;
    }
    if (PyList_GET_SIZE(self->f_emit_queue) > 0 &&
        !self->f_emit_scheduled)
        return fsm_schedule_flush(self);
    return 0;
}

PyObject *
FSM_flush_state_changed(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    if (fsm_flush_core((FSMOb *)self_) < 0)
        return NULL;
    Py_RETURN_NONE;
}

/* Batched flush: all FSMs that queue a stateChanged in the same loop
 * turn share one call_soon.  Relative order is preserved (FSMs drain in
 * first-queue order; each FSM drains its whole queue, exactly as the
 * per-FSM call_soon did).  While the batch is draining it stays
 * registered, so FSMs that queue *during* the drain (the serial
 * claim-chain case: a flush completes one claim cycle, whose callback
 * immediately starts the next claim) are appended to the same batch
 * and drained in the same loop callback.  That collapses long
 * callback-chained sequences into one event-loop turn instead of one
 * turn (epoll_wait + ready-queue bookkeeping) per link.  The drain is
 * capped per callback so a self-sustaining chain cannot starve IO; the
 * remainder is rescheduled with a fresh call_soon. */
#define FLUSH_DRAIN_CAP 48

typedef struct {
    PyObject_HEAD
    PyObject *fb_loop;
    PyObject *fb_list;   /* list of FSMOb */
    Py_ssize_t fb_pos;   /* next index to drain */
} FlushBatch;

extern PyTypeObject FlushBatchType;

static void
flushbatch_detach(FlushBatch *self)
{
    PyObject *cur = PyDict_GetItemWithError(g_flush_batches, self->fb_loop);
    if (cur == (PyObject *)self)
        (void)PyDict_DelItem(g_flush_batches, self->fb_loop);
    else if (cur == NULL)
        PyErr_Clear();
}

PyObject *
FlushBatch_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    FlushBatch *self = (FlushBatch *)self_;
    (void)args; (void)kwds;
    int rounds = 0;
    while (self->fb_pos < PyList_GET_SIZE(self->fb_list) &&
           rounds < FLUSH_DRAIN_CAP) {
        FSMOb *fsm = (FSMOb *)PyList_GET_ITEM(self->fb_list,
                                              self->fb_pos);
        self->fb_pos++;
        rounds++;
        if (fsm_flush_core(fsm) < 0) {
            /* un-wedge the FSMs we did not get to: their next
             * emission must be able to schedule a flush again */
            for (Py_ssize_t i = self->fb_pos;
                 i < PyList_GET_SIZE(self->fb_list); i++) {
                FSMOb *f = (FSMOb *)PyList_GET_ITEM(self->fb_list, i);
                f->f_emit_scheduled = 0;
            }
            flushbatch_detach(self);
            return NULL;
        }
    }
    if (self->fb_pos < PyList_GET_SIZE(self->fb_list)) {
        /* cap hit: yield to the loop, keep the batch registered so
         * new queuers keep appending, and finish on the next turn.
         * Compact the drained prefix first — under sustained load the
         * batch never empties and would otherwise grow without bound. */
        if (PyList_SetSlice(self->fb_list, 0, self->fb_pos, NULL) < 0) {
            flushbatch_detach(self);
            return NULL;
        }
        self->fb_pos = 0;
        PyObject *h = PyObject_CallMethodObjArgs(
            self->fb_loop, s_call_soon, self_, NULL);
        if (h == NULL) {
            flushbatch_detach(self);
            return NULL;
        }
        Py_DECREF(h);
        Py_RETURN_NONE;
    }
    flushbatch_detach(self);
    Py_RETURN_NONE;
}

int
FlushBatch_traverse(PyObject *self_, visitproc visit, void *arg)
{
    FlushBatch *self = (FlushBatch *)self_;
    Py_VISIT(self->fb_loop);
    Py_VISIT(self->fb_list);
    return 0;
}

int
FlushBatch_clear_(PyObject *self_)
{
    FlushBatch *self = (FlushBatch *)self_;
    Py_CLEAR(self->fb_loop);
    Py_CLEAR(self->fb_list);
    return 0;
}

void
FlushBatch_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    FlushBatch_clear_(self_);
    PyObject_GC_Del(self_);
}

PyTypeObject FlushBatchType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._FlushBatch",
    sizeof(FlushBatch),
    0,
    FlushBatch_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    FlushBatch_call,
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    0,
    FlushBatch_traverse,
    FlushBatch_clear_,
};

int
fsm_queue_state_changed(FSMOb *self, PyObject *state)
{
    if (PyList_Append(self->f_emit_queue, state) < 0)
        return -1;
    return fsm_schedule_flush(self);
}

int
fsm_schedule_flush(FSMOb *self)
{
    if (self->f_emit_scheduled)
        return 0;
    self->f_emit_scheduled = 1;
    PyObject *batch = PyDict_GetItemWithError(g_flush_batches, self->f_loop);
    if (batch == NULL) {
        if (PyErr_Occurred())
            return -1;
        FlushBatch *fb = PyObject_GC_New(FlushBatch, &FlushBatchType);
        if (fb == NULL)
            return -1;
        Py_INCREF(self->f_loop);
        fb->fb_loop = self->f_loop;
        fb->fb_pos = 0;
        fb->fb_list = PyList_New(0);
        PyObject_GC_Track((PyObject *)fb);
        if (fb->fb_list == NULL) {
            Py_DECREF(fb);
            return -1;
        }
        if (PyDict_SetItem(g_flush_batches, self->f_loop,
                           (PyObject *)fb) < 0) {
            Py_DECREF(fb);
            return -1;
        }
        PyObject *h = PyObject_CallMethodObjArgs(
            self->f_loop, s_call_soon, (PyObject *)fb, NULL);
        if (h == NULL) {
            Py_DECREF(fb);
            return -1;
        }
        Py_DECREF(h);
        batch = (PyObject *)fb;
        Py_DECREF(fb);  /* dict holds it */
    }
    if (PyList_Append(((FlushBatch *)batch)->fb_list,
                      (PyObject *)self) < 0)
        return -1;
    return 0;
}

/* resolve "state_x_y" attr name for a state string, cached globally */
PyObject *
entry_attr_name(PyObject *state)
{
    PyObject *name = PyDict_GetItemWithError(g_entry_name_cache, state);
    if (name != NULL) {
        Py_INCREF(name);
        return name;
    }
    if (PyErr_Occurred())
        return NULL;
    PyObject *replaced = PyObject_CallMethod(state, "replace", "OO",
                                             s_dot, s_underscore);
    if (replaced == NULL)
        return NULL;
    name = PyUnicode_Concat(s_state_prefix, replaced);
    Py_DECREF(replaced);
    if (name == NULL)
        return NULL;
    PyUnicode_InternInPlace(&name);
    if (PyDict_SetItem(g_entry_name_cache, state, name) < 0) {
        Py_DECREF(name);
        return NULL;
    }
    return name;
}

int
check_valid(FSMOb *self, PyObject *from, PyObject *state)
{
    if (self->f_valid == NULL)
        return 0;
    int c = PySequence_Contains(self->f_valid, state);
    if (c < 0)
        return -1;
    if (!c) {
        PyErr_Format(g_fsm_error,
                     "%s: invalid transition %R -> %R (valid: %R)",
                     Py_TYPE(self)->tp_name,
                     from ? from : Py_None, state, self->f_valid);
        return -1;
    }
    return 0;
}

int
fsm_enter_loop(FSMOb *self, PyObject *state)
{
    PyObject *next_state = state;
    Py_INCREF(next_state);
    while (next_state != NULL) {
        PyObject *target = next_state;
        next_state = NULL;
        if (self->f_scope != NULL) {
            PyObject *old = self->f_scope;
            self->f_scope = NULL;
            PyObject *r = Scope_dispose(old, NULL);
            Py_DECREF(old);
            if (r == NULL) {
                Py_DECREF(target);
                return -1;
            }
            Py_DECREF(r);
        }
        Py_CLEAR(self->f_valid);
        Py_INCREF(target);
        Py_XSETREF(self->f_state, target);
        if (PyList_Append(self->f_history, target) < 0) {
            Py_DECREF(target);
            return -1;
        }
        if (PyList_GET_SIZE(self->f_history) > 8) {
            if (PyList_SetSlice(self->f_history, 0, 1, NULL) < 0) {
                Py_DECREF(target);
                return -1;
            }
        }
        Scope *scope = PyObject_GC_New(Scope, &ScopeType);
        if (scope == NULL) {
            Py_DECREF(target);
            return -1;
        }
        Py_INCREF((PyObject *)self);
        scope->sc_fsm = (PyObject *)self;
        scope->sc_listeners = NULL;
        scope->sc_timers = NULL;
        scope->sc_active = 1;
        PyObject_GC_Track((PyObject *)scope);
        Py_INCREF((PyObject *)scope);
        Py_XSETREF(self->f_scope, (PyObject *)scope);

        if (g_tracer != NULL && g_tracer != Py_None) {
            PyObject *targs[2] = {(PyObject *)self, target};
            PyObject *tr = PyObject_Vectorcall(g_tracer, targs, 2, NULL);
            if (tr == NULL) {
                Py_DECREF(scope);
                Py_DECREF(target);
                return -1;
            }
            Py_DECREF(tr);
        }

        int handled = 0;
        if (self->f_fastkit != NULL) {
            self->f_entering = 1;
            handled = slotkit_maybe_entry(self->f_fastkit, self, target,
                                          scope);
            self->f_entering = 0;
            if (handled < 0) {
                Py_DECREF(scope);
                Py_DECREF(target);
                return -1;
            }
        }
        PyObject *r;
        if (handled) {
            Py_DECREF(scope);
            r = Py_None;
            Py_INCREF(r);
        } else {
            PyObject *attr = entry_attr_name(target);
            if (attr == NULL) {
                Py_DECREF(scope);
                Py_DECREF(target);
                return -1;
            }
            PyObject *entry = PyObject_GetAttr((PyObject *)self, attr);
            Py_DECREF(attr);
            if (entry == NULL) {
                PyErr_Clear();
                PyErr_Format(g_fsm_error,
                             "%s has no state-entry function for %R",
                             Py_TYPE(self)->tp_name, target);
                Py_DECREF(scope);
                Py_DECREF(target);
                return -1;
            }

            self->f_entering = 1;
            r = PyObject_CallOneArg(entry, (PyObject *)scope);
            Py_DECREF(entry);
            Py_DECREF(scope);
            self->f_entering = 0;
        }
        PyObject *pend = self->f_pending;
        self->f_pending = NULL;
        if (r == NULL) {
            Py_XDECREF(pend);
            Py_DECREF(target);
            return -1;
        }
        Py_DECREF(r);
        if (fsm_queue_state_changed(self, target) < 0) {
            Py_XDECREF(pend);
            Py_DECREF(target);
            return -1;
        }
        if (pend != NULL) {
            if (check_valid(self, target, pend) < 0) {
                Py_DECREF(pend);
                Py_DECREF(target);
                return -1;
            }
            next_state = pend;
        }
        Py_DECREF(target);
    }
    /* Settled in a claim-handle terminal state (they mark themselves
     * with the shared empty valid-transitions tuple): break the
     * per-claim reference cycles — fsm<->scope and handle<->ticket —
     * so the whole claim's object graph is freed by reference
     * counting.  Without this every claim leaves cyclic garbage and
     * CPython's cyclic GC pauses dominate claim p99 (measured 4x). */
    if (self->f_valid == t_empty &&
        PyObject_TypeCheck((PyObject *)self, &CHType))
        ch_terminal_cleanup(self);
    return 0;
}

int
fsm_goto_state(FSMOb *self, PyObject *state)
{
    if (check_valid(self, self->f_state, state) < 0)
        return -1;
    if (self->f_entering) {
        if (self->f_pending != NULL) {
            int eq = PyObject_RichCompareBool(self->f_pending, state, Py_EQ);
            if (eq < 0)
                return -1;
            if (!eq) {
                PyErr_Format(g_fsm_error,
                             "%s: conflicting deferred transitions %R and "
                             "%R from %R", Py_TYPE(self)->tp_name,
                             self->f_pending, state,
                             self->f_state ? self->f_state : Py_None);
                return -1;
            }
            return 0;
        }
        Py_INCREF(state);
        self->f_pending = state;
        return 0;
    }
    return fsm_enter_loop(self, state);
}

int
FSM_init(PyObject *self_, PyObject *args, PyObject *kwds)
{
    FSMOb *self = (FSMOb *)self_;
    PyObject *initial = NULL, *loop = Py_None;
    static const char *kwlist[] = {"initial_state", "loop", NULL};
    if (!PyArg_ParseTupleAndKeywords(args, kwds, "U|O",
                                     const_cast<char **>(kwlist),
                                     &initial, &loop))
        return -1;
    if (Emitter_init(self_, NULL, NULL) < 0)
        return -1;
    PyObject *resolved = PyObject_CallOneArg(g_get_loop, loop);
    if (resolved == NULL)
        return -1;
    Py_XSETREF(self->f_loop, resolved);
    if (self->f_emit_queue == NULL) {
        self->f_emit_queue = PyList_New(0);
        if (self->f_emit_queue == NULL)
            return -1;
    }
    if (self->f_history == NULL) {
        self->f_history = PyList_New(0);
        if (self->f_history == NULL)
            return -1;
    }
    Py_CLEAR(self->f_state);
    Py_CLEAR(self->f_valid);
    Py_CLEAR(self->f_pending);
    self->f_entering = 0;
    return fsm_goto_state(self, initial);
}

PyObject *
FSM_get_state(PyObject *self_, PyObject *noargs)
{
    FSMOb *self = (FSMOb *)self_;
    (void)noargs;
    if (self->f_state == NULL) {
        PyErr_SetString(g_fsm_error, "FSM has no state yet");
        return NULL;
    }
    Py_INCREF(self->f_state);
    return self->f_state;
}

PyObject *
FSM_is_in_state(PyObject *self_, PyObject *state)
{
    FSMOb *self = (FSMOb *)self_;
    if (self->f_state == NULL)
        Py_RETURN_FALSE;
    int eq = PyUnicode_Compare(self->f_state, state);
    if (eq == -1 && PyErr_Occurred())
        return NULL;
    if (eq == 0)
        Py_RETURN_TRUE;
    /* prefix semantics: cur startswith state + "." */
    Py_ssize_t sl = PyUnicode_GET_LENGTH(state);
    Py_ssize_t cl = PyUnicode_GET_LENGTH(self->f_state);
    if (cl > sl) {
        int m = PyUnicode_Tailmatch(self->f_state, state, 0, sl, -1);
        if (m < 0)
            return NULL;
        if (m && PyUnicode_ReadChar(self->f_state, sl) == '.')
            Py_RETURN_TRUE;
    }
    Py_RETURN_FALSE;
}

PyObject *
FSM_get_state_history(PyObject *self_, PyObject *noargs)
{
    FSMOb *self = (FSMOb *)self_;
    (void)noargs;
    return PyList_GetSlice(self->f_history, 0,
                           PyList_GET_SIZE(self->f_history));
}

PyObject *
FSM_goto_state_py(PyObject *self_, PyObject *state)
{
    if (!PyUnicode_Check(state)) {
        PyErr_SetString(PyExc_TypeError, "state must be a str");
        return NULL;
    }
    if (fsm_goto_state((FSMOb *)self_, state) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
FSM_get__fsm_state(PyObject *self_, void *closure)
{
    FSMOb *self = (FSMOb *)self_;
    (void)closure;
    if (self->f_state == NULL)
        Py_RETURN_NONE;
    Py_INCREF(self->f_state);
    return self->f_state;
}

PyObject *
FSM_get__fsm_valid(PyObject *self_, void *closure)
{
    FSMOb *self = (FSMOb *)self_;
    (void)closure;
    if (self->f_valid == NULL)
        Py_RETURN_NONE;
    Py_INCREF(self->f_valid);
    return self->f_valid;
}

int
FSM_set__fsm_valid(PyObject *self_, PyObject *value, void *closure)
{
    FSMOb *self = (FSMOb *)self_;
    (void)closure;
    if (value == NULL || value == Py_None) {
        Py_CLEAR(self->f_valid);
        return 0;
    }
    Py_INCREF(value);
    Py_XSETREF(self->f_valid, value);
    return 0;
}

PyObject *
Scope_valid_transitions(PyObject *self_, PyObject *states)
{
    Scope *self = (Scope *)self_;
    FSMOb *fsm = (FSMOb *)self->sc_fsm;
    Py_INCREF(states);
    Py_XSETREF(fsm->f_valid, states);
    Py_RETURN_NONE;
}

/* S.interval implemented natively with a tiny repeating driver */
typedef struct {
    PyObject_HEAD
    PyObject *iv_scope;
    PyObject *iv_cb;
    PyObject *iv_secs;
    PyObject *iv_handle;   /* current timer handle */
    int iv_stopped;
} IntervalOb;

extern PyTypeObject IntervalType;

PyObject *
Interval_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    IntervalOb *self = (IntervalOb *)self_;
    (void)args; (void)kwds;
    if (self->iv_stopped || !((Scope *)self->iv_scope)->sc_active)
        Py_RETURN_NONE;
    PyObject *r = PyObject_CallNoArgs(self->iv_cb);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);
    if (!self->iv_stopped && ((Scope *)self->iv_scope)->sc_active) {
        FSMOb *fsm = (FSMOb *)((Scope *)self->iv_scope)->sc_fsm;
        PyObject *h = PyObject_CallMethodObjArgs(
            fsm->f_loop, s_call_later, self->iv_secs, self_, NULL);
        if (h == NULL)
            return NULL;
        Py_XSETREF(self->iv_handle, h);
    }
    Py_RETURN_NONE;
}

PyObject *
Interval_cancel(PyObject *self_, PyObject *noargs)
{
    IntervalOb *self = (IntervalOb *)self_;
    (void)noargs;
    self->iv_stopped = 1;
    if (self->iv_handle != NULL) {
        PyObject *r = PyObject_CallMethodObjArgs(self->iv_handle, s_cancel,
                                                 NULL);
        if (r == NULL)
            return NULL;
        Py_DECREF(r);
    }
    Py_RETURN_NONE;
}

int
Interval_traverse(PyObject *self_, visitproc visit, void *arg)
{
    IntervalOb *self = (IntervalOb *)self_;
    Py_VISIT(self->iv_scope);
    Py_VISIT(self->iv_cb);
    Py_VISIT(self->iv_handle);
    return 0;
}

int
Interval_clear_(PyObject *self_)
{
    IntervalOb *self = (IntervalOb *)self_;
    Py_CLEAR(self->iv_scope);
    Py_CLEAR(self->iv_cb);
    Py_CLEAR(self->iv_secs);
    Py_CLEAR(self->iv_handle);
    return 0;
}

void
Interval_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    Interval_clear_(self_);
    PyObject_GC_Del(self_);
}

PyMethodDef Interval_methods[] = {
    {"cancel", Interval_cancel, METH_NOARGS, NULL},
    {NULL, NULL, 0, NULL},
};

PyTypeObject IntervalType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._Interval",
    sizeof(IntervalOb),
    0,
    Interval_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    Interval_call,
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    0,
    Interval_traverse,
    Interval_clear_,
    0, 0, 0, 0,
    Interval_methods,
};

PyObject *
Scope_interval(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    Scope *self = (Scope *)self_;
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "S.interval(ms, cb)");
        return NULL;
    }
    if (scope_check_active(self, "S.interval()") < 0)
        return NULL;
    double ms = PyFloat_AsDouble(args[0]);
    if (ms == -1.0 && PyErr_Occurred())
        return NULL;
    IntervalOb *iv = PyObject_GC_New(IntervalOb, &IntervalType);
    if (iv == NULL)
        return NULL;
    Py_INCREF(self_);
    iv->iv_scope = self_;
    Py_INCREF(args[1]);
    iv->iv_cb = args[1];
    iv->iv_secs = PyFloat_FromDouble(ms / 1000.0);
    iv->iv_handle = NULL;
    iv->iv_stopped = 0;
    PyObject_GC_Track((PyObject *)iv);
    if (iv->iv_secs == NULL) {
        Py_DECREF(iv);
        return NULL;
    }
    FSMOb *fsm = (FSMOb *)self->sc_fsm;
    PyObject *h = PyObject_CallMethodObjArgs(fsm->f_loop, s_call_later,
                                             iv->iv_secs, (PyObject *)iv,
                                             NULL);
    if (h == NULL) {
        Py_DECREF(iv);
        return NULL;
    }
    iv->iv_handle = h;
    int rc = scope_add_timer(self, (PyObject *)iv);
    Py_DECREF(iv);
    if (rc < 0)
        return NULL;
    Py_RETURN_NONE;
}

int
FSM_traverse(PyObject *self_, visitproc visit, void *arg)
{
    FSMOb *self = (FSMOb *)self_;
    Py_VISIT(self->f_loop);
    Py_VISIT(self->f_state);
    Py_VISIT(self->f_scope);
    Py_VISIT(self->f_valid);
    Py_VISIT(self->f_pending);
    Py_VISIT(self->f_emit_queue);
    Py_VISIT(self->f_history);
    Py_VISIT(self->f_flush_bound);
    Py_VISIT(self->f_fastkit);
    return Emitter_traverse(self_, visit, arg);
}

int
FSM_clear_(PyObject *self_)
{
    FSMOb *self = (FSMOb *)self_;
    Py_CLEAR(self->f_loop);
    Py_CLEAR(self->f_state);
    Py_CLEAR(self->f_scope);
    Py_CLEAR(self->f_valid);
    Py_CLEAR(self->f_pending);
    Py_CLEAR(self->f_emit_queue);
    Py_CLEAR(self->f_history);
    Py_CLEAR(self->f_flush_bound);
    Py_CLEAR(self->f_fastkit);
    return Emitter_clear_(self_);
}

void
FSM_dealloc(PyObject *self_)
{
    FSMOb *self = (FSMOb *)self_;
    PyTypeObject *tp = Py_TYPE(self_);
    PyObject_GC_UnTrack(self_);
    if (self->base.ev_weakrefs != NULL)
        PyObject_ClearWeakRefs(self_);
    FSM_clear_(self_);
    tp->tp_free(self_);
}

static PyObject *
FSM_set_fast_kit(PyObject *self_, PyObject *kit)
{
    FSMOb *self = (FSMOb *)self_;
    if (kit == Py_None) {
        Py_CLEAR(self->f_fastkit);
    } else {
        Py_INCREF(kit);
        Py_XSETREF(self->f_fastkit, kit);
    }
    Py_RETURN_NONE;
}

PyMethodDef FSM_methods[] = {
    {"get_state", FSM_get_state, METH_NOARGS, NULL},
    {"is_in_state", FSM_is_in_state, METH_O, NULL},
    {"get_state_history", FSM_get_state_history, METH_NOARGS, NULL},
    {"goto_state", FSM_goto_state_py, METH_O, NULL},
    {"_flush_state_changed", FSM_flush_state_changed, METH_NOARGS, NULL},
    {"_set_fast_kit", FSM_set_fast_kit, METH_O, NULL},
    {NULL, NULL, 0, NULL},
};

PyMemberDef FSM_members[] = {
    {(char *)"_loop", T_OBJECT, offsetof(FSMOb, f_loop), READONLY, NULL},
    {NULL, 0, 0, 0, NULL},
};

PyGetSetDef FSM_getset[] = {
    {(char *)"_fsm_state", FSM_get__fsm_state, NULL, NULL, NULL},
    {(char *)"_fsm_valid", FSM_get__fsm_valid, FSM_set__fsm_valid, NULL,
     NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject FSMType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.FSM",
    sizeof(FSMOb),
    0,
    FSM_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_BASETYPE | Py_TPFLAGS_HAVE_GC,
    "Moore machine (native)",
    FSM_traverse,
    FSM_clear_,
    0, 0, 0, 0,
    FSM_methods,
    FSM_members,
    FSM_getset,
    &EmitterType,                         /* tp_base */
    0, 0, 0,
    0,                                    /* tp_dictoffset (inherited) */
    FSM_init,
    0,
    PyType_GenericNew,
};


/* ------------------------------------------------------------------ */
/* ClaimHandleBase: the pool claim handle's hot path in C              */
/* ------------------------------------------------------------------ */

PyObject *g_err_claim_timeout;   /* errors.ClaimTimeoutError */
PyObject *g_err_cueball;         /* errors.CueballError */
PyObject *g_err_misused;         /* errors.ClaimHandleMisusedError */
PyObject *g_capture_stack;       /* utils.maybe_capture_stack_trace */

PyObject *s_claim;               /* "claim" */
PyObject *s_warn;                /* "warn" */
PyObject *s_incr;                /* "_incr_counter" */
PyObject *s_claim_timeout_evt;   /* "claim-timeout" */
PyObject *s_is_in_state;         /* "is_in_state" */
PyObject *s_time;                /* "time" */
PyObject *s_waiting, *s_claiming, *s_claimed, *s_released, *s_closed,
         *s_cancelled, *s_failed, *s_idle, *s_error_evt;
PyObject *t_waiting_valid;       /* ("claiming","cancelled","failed") */
PyObject *t_claiming_valid;      /* ("claimed","waiting","cancelled") */
PyObject *t_claimed_valid;       /* ("released","closed") */
PyObject *t_empty;               /* () */
PyObject *s_leak_events[4];      /* close, error, readable, data */

/* defined in the SlotKit section below */
extern PyObject *s_busy_st;
extern PyObject *s_csf_handle;

typedef struct {
    FSMOb base;
    PyObject *chb_pool;
    PyObject *chb_claim_stack;
    PyObject *chb_callback;
    PyObject *chb_log;
    PyObject *chb_slot;
    PyObject *chb_release_stack;
    PyObject *chb_connection;
    PyObject *chb_last_error;
    double chb_claim_timeout;
    double chb_started;
    long chb_pre[4];
    int chb_have_pre;
    int chb_throw_error;
    int chb_cancelled;
    int chb_leak_check;
    int chb_pinger;
} CHOb;

extern PyTypeObject CHType;

long
c_count_listeners(PyObject *emitter, PyObject *event)
{
    /* native-emitter path of count_listeners (callers guarantee the
     * emitter type or fall back in python) */
    if (!PyObject_TypeCheck(emitter, &EmitterType))
        return -2;  /* not ours: caller must use python fallback */
    Emitter *em = (Emitter *)emitter;
    if (em->ev_events == NULL)
        return 0;
    PyObject *ls = PyDict_GetItemWithError(em->ev_events, event);
    if (ls == NULL)
        return PyErr_Occurred() ? -1 : 0;
    long count = 0;
    Py_ssize_t n = PyList_GET_SIZE(ls);
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *h = PyList_GET_ITEM(ls, i);
        if (!PyCallable_Check(h))
            continue;
        PyObject *marker = NULL;
        if (_PyObject_LookupAttr(h, s_internal, &marker) < 0)
            return -1;
        if (marker != NULL) {
            int truthy = PyObject_IsTrue(marker);
            Py_DECREF(marker);
            if (truthy < 0)
                return -1;
            if (truthy)
                continue;
        }
        PyObject *target = NULL;
        if (Py_TYPE(h) == &OnceWrapperType) {
            target = ((OnceWrapper *)h)->ow_listener;
            Py_INCREF(target);
        } else if (_PyObject_LookupAttr(h, s_listener, &target) < 0) {
            return -1;
        }
        if (target != NULL && target != h) {
            if (_PyObject_LookupAttr(target, s_internal, &marker) < 0) {
                Py_DECREF(target);
                return -1;
            }
            if (marker != NULL) {
                int truthy = PyObject_IsTrue(marker);
                Py_DECREF(marker);
                if (truthy < 0) {
                    Py_DECREF(target);
                    return -1;
                }
                if (truthy) {
                    Py_DECREF(target);
                    continue;
                }
            }
        }
        Py_XDECREF(target);
        count++;
    }
    return count;
}

/* claimed-state conn-error listener: raise if the user has no error
 * listener (and throw_error), else warn + count */
typedef struct {
    PyObject_HEAD
    PyObject *ce_handle;  /* CHOb */
} ConnErrCb;

extern PyTypeObject ConnErrCbType;

PyObject *
ConnErrCb_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    CHOb *h = (CHOb *)((ConnErrCb *)self_)->ce_handle;
    (void)kwds;
    PyObject *err = (PyTuple_GET_SIZE(args) > 0)
        ? PyTuple_GET_ITEM(args, 0) : Py_None;
    long cnt = c_count_listeners(h->chb_connection, s_error_evt);
    if (cnt == -1)
        return NULL;
    if (cnt == 0 && h->chb_throw_error) {
        /* end-user registered no 'error' listener: surface loudly
         * (lib/connection-fsm.js:697-706) */
        if (PyExceptionInstance_Check(err)) {
            PyErr_SetObject((PyObject *)Py_TYPE(err), err);
        } else {
            PyErr_SetObject(PyExc_RuntimeError, err);
        }
        return NULL;
    }
    PyObject *msg = PyUnicode_FromString(
        "connection emitted error while claimed");
    if (msg == NULL)
        return NULL;
    PyObject *r = PyObject_CallMethodObjArgs(h->chb_log, s_warn, msg,
                                             NULL);
    Py_DECREF(msg);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);
    PyObject *evt = PyUnicode_FromString("error-while-claimed");
    if (evt == NULL)
        return NULL;
    r = PyObject_CallMethodObjArgs(h->chb_pool, s_incr, evt, NULL);
    Py_DECREF(evt);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);
    Py_RETURN_NONE;
}

int
ConnErrCb_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Py_VISIT(((ConnErrCb *)self_)->ce_handle);
    return 0;
}

int
ConnErrCb_clear_(PyObject *self_)
{
    Py_CLEAR(((ConnErrCb *)self_)->ce_handle);
    return 0;
}

void
ConnErrCb_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    ConnErrCb_clear_(self_);
    PyObject_GC_Del(self_);
}

PyObject *
ConnErrCb_get_internal(PyObject *self_, void *closure)
{
    (void)self_; (void)closure;
    Py_RETURN_TRUE;   /* _cueball_internal marker */
}

PyGetSetDef ConnErrCb_getset[] = {
    {(char *)"_cueball_internal", ConnErrCb_get_internal, NULL, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject ConnErrCbType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._ConnErrCb",
    sizeof(ConnErrCb),
    0,
    ConnErrCb_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    ConnErrCb_call,
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    0,
    ConnErrCb_traverse,
    ConnErrCb_clear_,
    0, 0, 0, 0, 0, 0,
    ConnErrCb_getset,
};

/* failed-state deferred callback: cb(last_error) */
typedef struct {
    PyObject_HEAD
    PyObject *fc_handle;  /* CHOb */
} FailCb;

extern PyTypeObject FailCbType;

PyObject *
FailCb_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    CHOb *h = (CHOb *)((FailCb *)self_)->fc_handle;
    (void)args; (void)kwds;
    PyObject *err = h->chb_last_error ? h->chb_last_error : Py_None;
    return PyObject_CallOneArg(h->chb_callback, err);
}

int
FailCb_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Py_VISIT(((FailCb *)self_)->fc_handle);
    return 0;
}

int
FailCb_clear_(PyObject *self_)
{
    Py_CLEAR(((FailCb *)self_)->fc_handle);
    return 0;
}

void
FailCb_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    FailCb_clear_(self_);
    PyObject_GC_Del(self_);
}

PyTypeObject FailCbType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._FailCb",
    sizeof(FailCb),
    0,
    FailCb_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0,
    FailCb_call,
    0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC,
    0,
    FailCb_traverse,
    FailCb_clear_,
};

/* -- CH helpers ------------------------------------------------------- */

int
ch_state_is(CHOb *self, PyObject *state)
{
    return self->base.f_state == state ||
        (self->base.f_state != NULL &&
         PyUnicode_Compare(self->base.f_state, state) == 0);
}

PyObject *
ch_err_format(PyObject *cls, const char *msg)
{
    PyErr_SetString(cls, msg);
    return NULL;
}

/* -- signal methods ---------------------------------------------------- */

PyObject *
CH_try_(PyObject *self_, PyObject *slot)
{
    CHOb *self = (CHOb *)self_;
    if (!ch_state_is(self, s_waiting))
        return ch_err_format(g_fsm_error,
            "ClaimHandle.try_ only in \"waiting\"");
    PyObject *r = PyObject_CallMethodObjArgs(slot, s_is_in_state, s_idle,
                                             NULL);
    if (r == NULL)
        return NULL;
    int idle = PyObject_IsTrue(r);
    Py_DECREF(r);
    if (!idle)
        return ch_err_format(g_fsm_error,
            "ClaimHandle.try_ needs an idle slot");
    Py_INCREF(slot);
    Py_XSETREF(self->chb_slot, slot);
    if (fsm_goto_state(&self->base, s_claiming) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
CH_accept(PyObject *self_, PyObject *connection)
{
    CHOb *self = (CHOb *)self_;
    if (!ch_state_is(self, s_claiming))
        return ch_err_format(g_fsm_error, "accept only in claiming");
    Py_INCREF(connection);
    Py_XSETREF(self->chb_connection, connection);
    if (fsm_goto_state(&self->base, s_claimed) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
CH_reject(PyObject *self_, PyObject *noargs)
{
    CHOb *self = (CHOb *)self_;
    (void)noargs;
    if (!ch_state_is(self, s_claiming))
        return ch_err_format(g_fsm_error, "reject only in claiming");
    if (fsm_goto_state(&self->base,
                       self->chb_cancelled ? s_cancelled : s_waiting) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *CH_release(PyObject *self_, PyObject *noargs);

PyObject *
CH_cancel(PyObject *self_, PyObject *noargs)
{
    CHOb *self = (CHOb *)self_;
    if (ch_state_is(self, s_claimed))
        return CH_release(self_, noargs);
    self->chb_cancelled = 1;
    /* in "claiming" cancellation applies on reject/accept; in
     * "waiting" it takes effect now (lib/connection-fsm.js:580) */
    if (ch_state_is(self, s_waiting)) {
        if (fsm_goto_state(&self->base, s_cancelled) < 0)
            return NULL;
    }
    Py_RETURN_NONE;
}

int
ch_do_claim_timeout(CHOb *self)
{
    PyObject *err = PyObject_CallOneArg(g_err_claim_timeout,
                                        self->chb_pool);
    if (err == NULL)
        return -1;
    Py_XSETREF(self->chb_last_error, err);
    PyObject *r = PyObject_CallMethodObjArgs(
        self->chb_pool, s_incr, s_claim_timeout_evt, NULL);
    if (r == NULL)
        return -1;
    Py_DECREF(r);
    return fsm_goto_state(&self->base, s_failed);
}

PyObject *
CH_timeout(PyObject *self_, PyObject *noargs)
{
    CHOb *self = (CHOb *)self_;
    (void)noargs;
    if (!ch_state_is(self, s_waiting))
        return ch_err_format(g_fsm_error, "timeout only in waiting");
    if (ch_do_claim_timeout(self) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
CH__on_claim_timeout(PyObject *self_, PyObject *noargs)
{
    return CH_timeout(self_, noargs);
}

PyObject *
CH_fail(PyObject *self_, PyObject *err)
{
    CHOb *self = (CHOb *)self_;
    if (ch_state_is(self, s_waiting)) {
        Py_INCREF(err);
        Py_XSETREF(self->chb_last_error, err);
        if (fsm_goto_state(&self->base, s_failed) < 0)
            return NULL;
    }
    Py_RETURN_NONE;
}

int
ch_relinquish(CHOb *self, PyObject *target_state)
{
    if (!ch_state_is(self, s_claimed)) {
        if (ch_state_is(self, s_released) || ch_state_is(self, s_closed)) {
            /* capture ran from C, so the innermost python frame IS
             * the release() caller (the pure-python twin captures two
             * frames deeper and indexes -3) */
            PyObject *by = NULL;
            PyObject *stack = self->chb_release_stack;
            if (stack != NULL && PyList_Check(stack) &&
                PyList_GET_SIZE(stack) >= 1) {
                by = PyList_GET_ITEM(stack, PyList_GET_SIZE(stack) - 1);
            }
            PyObject *msg = PyUnicode_FromFormat(
                "Connection not claimed by this handle, released by %S",
                by ? by : Py_None);
            if (msg == NULL)
                return -1;
            PyObject *err = PyObject_CallOneArg(g_err_cueball, msg);
            Py_DECREF(msg);
            if (err == NULL)
                return -1;
            PyErr_SetObject(g_err_cueball, err);
            Py_DECREF(err);
            return -1;
        }
        PyObject *msg = PyUnicode_FromFormat(
            "ClaimHandle.release() called while in state \"%S\"",
            self->base.f_state ? self->base.f_state : Py_None);
        if (msg == NULL)
            return -1;
        PyErr_SetObject(g_err_cueball, msg);
        Py_DECREF(msg);
        return -1;
    }
    PyObject *stack = PyObject_CallNoArgs(g_capture_stack);
    if (stack == NULL)
        return -1;
    Py_XSETREF(self->chb_release_stack, stack);
    return fsm_goto_state(&self->base, target_state);
}

PyObject *
CH_release(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    if (ch_relinquish((CHOb *)self_, s_released) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
CH_close(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    if (ch_relinquish((CHOb *)self_, s_closed) < 0)
        return NULL;
    Py_RETURN_NONE;
}

PyObject *
CH_disable_release_leak_check(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    ((CHOb *)self_)->chb_leak_check = 0;
    Py_RETURN_NONE;
}

/* -- state entries ------------------------------------------------------ */

PyObject *
CH_state_waiting(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    Py_INCREF(t_waiting_valid);
    Py_XSETREF(self->base.f_valid, t_waiting_valid);
    Py_CLEAR(self->chb_slot);
    if (isfinite(self->chb_claim_timeout)) {
        PyObject *ms = PyFloat_FromDouble(self->chb_claim_timeout);
        if (ms == NULL)
            return NULL;
        PyObject *cb = PyObject_GetAttrString(self_, "_on_claim_timeout");
        if (cb == NULL) {
            Py_DECREF(ms);
            return NULL;
        }
        PyObject *args[2] = {ms, cb};
        PyObject *r = Scope_timeout(scope, args, 2);
        Py_DECREF(ms);
        Py_DECREF(cb);
        if (r == NULL)
            return NULL;
        Py_DECREF(r);
    }
    Py_RETURN_NONE;
}

PyObject *
CH_state_claiming(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    (void)scope;
    Py_INCREF(t_claiming_valid);
    Py_XSETREF(self->base.f_valid, t_claiming_valid);
    PyObject *slot = self->chb_slot;
    /* inline of ConnectionSlotFSM.claim() when the slot runs the
     * native busy/idle cycle (same checks, same transition) */
    if (slot != NULL && PyObject_TypeCheck(slot, &FSMType) &&
        ((FSMOb *)slot)->f_fastkit != NULL) {
        FSMOb *sf = (FSMOb *)slot;
        PyObject *fs = sf->f_state;
        if (!(fs == s_idle ||
              (fs != NULL && PyUnicode_Compare(fs, s_idle) == 0))) {
            PyErr_SetString(g_fsm_error, "claim only in idle");
            return NULL;
        }
        PyObject *cur = PyObject_GetAttr(slot, s_csf_handle);
        if (cur == NULL)
            return NULL;
        int has = (cur != Py_None);
        Py_DECREF(cur);
        if (has) {
            PyErr_SetString(g_fsm_error, "slot already has a handle");
            return NULL;
        }
        if (PyObject_SetAttr(slot, s_csf_handle, self_) < 0)
            return NULL;
        if (fsm_goto_state(sf, s_busy_st) < 0)
            return NULL;
        Py_RETURN_NONE;
    }
    PyObject *r = PyObject_CallMethodObjArgs(slot, s_claim,
                                             self_, NULL);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);
    Py_RETURN_NONE;
}

PyObject *
CH_state_claimed(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    Py_INCREF(t_claimed_valid);
    Py_XSETREF(self->base.f_valid, t_claimed_valid);

    if (self->chb_cancelled) {
        if (fsm_goto_state(&self->base, s_released) < 0)
            return NULL;
        Py_RETURN_NONE;
    }

    PyObject *conn = self->chb_connection;
    for (int i = 0; i < 4; i++) {
        long c = c_count_listeners(conn, s_leak_events[i]);
        if (c == -1)
            return NULL;
        self->chb_pre[i] = (c == -2) ? 0 : c;
    }
    self->chb_have_pre = 1;

    ConnErrCb *ec = PyObject_GC_New(ConnErrCb, &ConnErrCbType);
    if (ec == NULL)
        return NULL;
    Py_INCREF(self_);
    ec->ce_handle = self_;
    PyObject_GC_Track((PyObject *)ec);
    PyObject *args[3] = {conn, s_error_evt, (PyObject *)ec};
    PyObject *r = Scope_on(scope, args, 3);
    Py_DECREF(ec);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);

    PyObject *cbargs[3] = {Py_None, self_, conn};
    r = PyObject_Vectorcall(self->chb_callback, cbargs, 3, NULL);
    if (r == NULL)
        return NULL;
    Py_DECREF(r);
    Py_RETURN_NONE;
}

PyObject *
CH_state_released(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    (void)scope;
    Py_INCREF(t_empty);
    Py_XSETREF(self->base.f_valid, t_empty);
    if (!self->chb_leak_check || !self->chb_have_pre)
        Py_RETURN_NONE;
    PyObject *conn = self->chb_connection;
    for (int i = 0; i < 4; i++) {
        long c = c_count_listeners(conn, s_leak_events[i]);
        if (c == -1)
            return NULL;
        if (c != -2 && c > self->chb_pre[i]) {
            PyObject *msg = PyUnicode_FromFormat(
                "connection claimer looks like it leaked event handlers "
                "(event=%S before=%ld after=%ld)",
                s_leak_events[i], self->chb_pre[i], c);
            if (msg == NULL)
                return NULL;
            PyObject *r = PyObject_CallMethodObjArgs(self->chb_log, s_warn,
                                                     msg, NULL);
            Py_DECREF(msg);
            if (r == NULL)
                return NULL;
            Py_DECREF(r);
        }
    }
    Py_RETURN_NONE;
}

PyObject *
CH_state_closed(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    (void)scope;
    Py_INCREF(t_empty);
    Py_XSETREF(self->base.f_valid, t_empty);
    Py_RETURN_NONE;
}

PyObject *
CH_state_cancelled(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    (void)scope;
    Py_INCREF(t_empty);
    Py_XSETREF(self->base.f_valid, t_empty);
    Py_RETURN_NONE;
}

PyObject *
CH_state_failed(PyObject *self_, PyObject *scope)
{
    CHOb *self = (CHOb *)self_;
    (void)scope;
    Py_INCREF(t_empty);
    Py_XSETREF(self->base.f_valid, t_empty);
    FailCb *fc = PyObject_GC_New(FailCb, &FailCbType);
    if (fc == NULL)
        return NULL;
    Py_INCREF(self_);
    fc->fc_handle = self_;
    PyObject_GC_Track((PyObject *)fc);
    /* schedule directly, not through the scope: failed is terminal so
     * the callback can never become stale, and keeping the terminal
     * scope empty lets ch_terminal_cleanup dispose it (cycle break) */
    PyObject *h = PyObject_CallMethodObjArgs(self->base.f_loop,
                                             s_call_soon,
                                             (PyObject *)fc, NULL);
    Py_DECREF(fc);
    if (h == NULL)
        return NULL;
    Py_DECREF(h);
    Py_RETURN_NONE;
}

/* -- misuse traps -------------------------------------------------------- */

PyObject *
CH_get_misused(PyObject *self_, void *closure)
{
    (void)self_; (void)closure;
    PyObject *err = PyObject_CallNoArgs(g_err_misused);
    if (err == NULL)
        return NULL;
    PyErr_SetObject(g_err_misused, err);
    Py_DECREF(err);
    return NULL;
}

PyObject *
CH_on(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs >= 1 && PyUnicode_Check(args[0])) {
        if (PyUnicode_CompareWithASCIIString(args[0], "readable") == 0 ||
            PyUnicode_CompareWithASCIIString(args[0], "close") == 0) {
            PyObject *err = PyObject_CallNoArgs(g_err_misused);
            if (err == NULL)
                return NULL;
            PyErr_SetObject(g_err_misused, err);
            Py_DECREF(err);
            return NULL;
        }
    }
    return Emitter_on(self_, args, nargs);
}

PyObject *
CH_once(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs >= 1 && PyUnicode_Check(args[0])) {
        if (PyUnicode_CompareWithASCIIString(args[0], "readable") == 0 ||
            PyUnicode_CompareWithASCIIString(args[0], "close") == 0) {
            PyObject *err = PyObject_CallNoArgs(g_err_misused);
            if (err == NULL)
                return NULL;
            PyErr_SetObject(g_err_misused, err);
            Py_DECREF(err);
            return NULL;
        }
    }
    return Emitter_once(self_, args, nargs);
}

/* -- init / gc ------------------------------------------------------------ */

int
CH_traverse(PyObject *self_, visitproc visit, void *arg)
{
    CHOb *self = (CHOb *)self_;
    Py_VISIT(self->chb_pool);
    Py_VISIT(self->chb_claim_stack);
    Py_VISIT(self->chb_callback);
    Py_VISIT(self->chb_log);
    Py_VISIT(self->chb_slot);
    Py_VISIT(self->chb_release_stack);
    Py_VISIT(self->chb_connection);
    Py_VISIT(self->chb_last_error);
    return FSM_traverse(self_, visit, arg);
}

int
CH_clear_(PyObject *self_)
{
    CHOb *self = (CHOb *)self_;
    Py_CLEAR(self->chb_pool);
    Py_CLEAR(self->chb_claim_stack);
    Py_CLEAR(self->chb_callback);
    Py_CLEAR(self->chb_log);
    Py_CLEAR(self->chb_slot);
    Py_CLEAR(self->chb_release_stack);
    Py_CLEAR(self->chb_connection);
    Py_CLEAR(self->chb_last_error);
    return FSM_clear_(self_);
}

void
CH_dealloc(PyObject *self_)
{
    CHOb *self = (CHOb *)self_;
    PyTypeObject *tp = Py_TYPE(self_);
    PyObject_GC_UnTrack(self_);
    if (self->base.base.ev_weakrefs != NULL)
        PyObject_ClearWeakRefs(self_);
    CH_clear_(self_);
    tp->tp_free(self_);
}

/* _setup(pool, claim_stack, callback, log, claim_timeout, loop):
 * one-shot C-side initializer used by the python subclass */
PyObject *
CH__setup(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    CHOb *self = (CHOb *)self_;
    if (nargs != 7) {
        PyErr_SetString(PyExc_TypeError,
            "_setup(pool, stack, cb, log, timeout, throw_error, loop)");
        return NULL;
    }
    Py_INCREF(args[0]); Py_XSETREF(self->chb_pool, args[0]);
    Py_INCREF(args[1]); Py_XSETREF(self->chb_claim_stack, args[1]);
    Py_INCREF(args[2]); Py_XSETREF(self->chb_callback, args[2]);
    Py_INCREF(args[3]); Py_XSETREF(self->chb_log, args[3]);
    self->chb_claim_timeout = PyFloat_AsDouble(args[4]);
    if (self->chb_claim_timeout == -1.0 && PyErr_Occurred())
        return NULL;
    int throw_error = PyObject_IsTrue(args[5]);
    if (throw_error < 0)
        return NULL;
    self->chb_throw_error = throw_error;
    self->chb_cancelled = 0;
    self->chb_leak_check = 1;
    self->chb_pinger = 0;
    self->chb_have_pre = 0;

    /* FSM init (resolves loop, enters "waiting") */
    PyObject *resolved = PyObject_CallOneArg(g_get_loop, args[6]);
    if (resolved == NULL)
        return NULL;
    Py_XSETREF(self->base.f_loop, resolved);
    if (Emitter_init(self_, NULL, NULL) < 0)
        return NULL;
    if (self->base.f_emit_queue == NULL) {
        self->base.f_emit_queue = PyList_New(0);
        if (self->base.f_emit_queue == NULL)
            return NULL;
    }
    if (self->base.f_history == NULL) {
        self->base.f_history = PyList_New(0);
        if (self->base.f_history == NULL)
            return NULL;
    }
    if (fsm_goto_state(&self->base, s_waiting) < 0)
        return NULL;
    PyObject *t = PyObject_CallMethodObjArgs(self->base.f_loop, s_time,
                                             NULL);
    if (t == NULL)
        return NULL;
    double secs = PyFloat_AsDouble(t);
    Py_DECREF(t);
    if (secs == -1.0 && PyErr_Occurred())
        return NULL;
    self->chb_started = secs * 1000.0;
    Py_RETURN_NONE;
}

PyMethodDef CH_methods[] = {
    {"_setup", (PyCFunction)(void (*)(void))CH__setup, METH_FASTCALL, NULL},
    {"try_", CH_try_, METH_O, NULL},
    {"accept", CH_accept, METH_O, NULL},
    {"reject", CH_reject, METH_NOARGS, NULL},
    {"cancel", CH_cancel, METH_NOARGS, NULL},
    {"timeout", CH_timeout, METH_NOARGS, NULL},
    {"_on_claim_timeout", CH__on_claim_timeout, METH_NOARGS, NULL},
    {"fail", CH_fail, METH_O, NULL},
    {"release", CH_release, METH_NOARGS, NULL},
    {"close", CH_close, METH_NOARGS, NULL},
    {"disable_release_leak_check", CH_disable_release_leak_check,
     METH_NOARGS, NULL},
    {"on", (PyCFunction)(void (*)(void))CH_on, METH_FASTCALL, NULL},
    {"once", (PyCFunction)(void (*)(void))CH_once, METH_FASTCALL, NULL},
    {"state_waiting", CH_state_waiting, METH_O, NULL},
    {"state_claiming", CH_state_claiming, METH_O, NULL},
    {"state_claimed", CH_state_claimed, METH_O, NULL},
    {"state_released", CH_state_released, METH_O, NULL},
    {"state_closed", CH_state_closed, METH_O, NULL},
    {"state_cancelled", CH_state_cancelled, METH_O, NULL},
    {"state_failed", CH_state_failed, METH_O, NULL},
    {NULL, NULL, 0, NULL},
};

PyMemberDef CH_members[] = {
    {(char *)"ch_pool", T_OBJECT, offsetof(CHOb, chb_pool), 0, NULL},
    {(char *)"ch_claim_stack", T_OBJECT, offsetof(CHOb, chb_claim_stack),
     0, NULL},
    {(char *)"ch_callback", T_OBJECT, offsetof(CHOb, chb_callback), 0,
     NULL},
    {(char *)"ch_log", T_OBJECT, offsetof(CHOb, chb_log), 0, NULL},
    {(char *)"ch_slot", T_OBJECT, offsetof(CHOb, chb_slot), 0, NULL},
    {(char *)"ch_release_stack", T_OBJECT,
     offsetof(CHOb, chb_release_stack), 0, NULL},
    {(char *)"ch_connection", T_OBJECT, offsetof(CHOb, chb_connection),
     0, NULL},
    {(char *)"ch_last_error", T_OBJECT, offsetof(CHOb, chb_last_error),
     0, NULL},
    {(char *)"ch_claim_timeout", T_DOUBLE,
     offsetof(CHOb, chb_claim_timeout), 0, NULL},
    {(char *)"ch_started", T_DOUBLE, offsetof(CHOb, chb_started), 0, NULL},
    {(char *)"ch_cancelled", T_INT, offsetof(CHOb, chb_cancelled), 0,
     NULL},
    {(char *)"ch_throw_error", T_INT, offsetof(CHOb, chb_throw_error), 0,
     NULL},
    {(char *)"ch_do_release_leak_check", T_INT,
     offsetof(CHOb, chb_leak_check), 0, NULL},
    {(char *)"ch_pinger", T_INT, offsetof(CHOb, chb_pinger), 0, NULL},
    {NULL, 0, 0, 0, NULL},
};

PyGetSetDef CH_getset[] = {
    {(char *)"readable", CH_get_misused, NULL, NULL, NULL},
    {(char *)"writable", CH_get_misused, NULL, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject CHType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.ClaimHandleBase",
    sizeof(CHOb),
    0,
    CH_dealloc,
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    Py_TPFLAGS_DEFAULT | Py_TPFLAGS_BASETYPE | Py_TPFLAGS_HAVE_GC,
    "pool claim handle (native core)",
    CH_traverse,
    CH_clear_,
    0, 0, 0, 0,
    CH_methods,
    CH_members,
    CH_getset,
    &FSMType,
    0, 0, 0,
    0,
    0,
    0,
    PyType_GenericNew,
};

PyObject *
speed_set_claim_helpers(PyObject *mod, PyObject *const *args,
                        Py_ssize_t nargs)
{
    (void)mod;
    if (nargs != 4) {
        PyErr_SetString(PyExc_TypeError,
            "_set_claim_helpers(ClaimTimeoutError, CueballError, "
            "ClaimHandleMisusedError, capture_stack)");
        return NULL;
    }
    Py_INCREF(args[0]); Py_XSETREF(g_err_claim_timeout, args[0]);
    Py_INCREF(args[1]); Py_XSETREF(g_err_cueball, args[1]);
    Py_INCREF(args[2]); Py_XSETREF(g_err_misused, args[2]);
    Py_INCREF(args[3]); Py_XSETREF(g_capture_stack, args[3]);
    Py_RETURN_NONE;
}

/* ------------------------------------------------------------------ */
/* module                                                              */
/* ------------------------------------------------------------------ */

PyObject *
speed_set_helpers(PyObject *mod, PyObject *const *args, Py_ssize_t nargs)
{
    (void)mod;
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "_set_helpers(get_loop, FSMError)");
        return NULL;
    }
    Py_INCREF(args[0]);
    Py_XSETREF(g_get_loop, args[0]);
    Py_INCREF(args[1]);
    Py_XSETREF(g_fsm_error, args[1]);
    Py_RETURN_NONE;
}

PyObject *
speed_count_listeners(PyObject *mod, PyObject *const *args,
                      Py_ssize_t nargs)
{
    (void)mod;
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "count_listeners(emitter, event)");
        return NULL;
    }
    if (!PyObject_TypeCheck(args[0], &EmitterType)) {
        PyErr_SetString(PyExc_TypeError, "emitter must be an EventEmitter");
        return NULL;
    }
    Emitter *em = (Emitter *)args[0];
    if (em->ev_events == NULL)
        return PyLong_FromLong(0);
    PyObject *ls = PyDict_GetItemWithError(em->ev_events, args[1]);
    if (ls == NULL)
        return PyErr_Occurred() ? NULL : PyLong_FromLong(0);
    Py_ssize_t n = PyList_GET_SIZE(ls);
    long count = 0;
    for (Py_ssize_t i = 0; i < n; i++) {
        PyObject *h = PyList_GET_ITEM(ls, i);
        if (!PyCallable_Check(h))
            continue;
        PyObject *marker = NULL;
        if (_PyObject_LookupAttr(h, s_internal, &marker) < 0)
            return NULL;
        if (marker != NULL) {
            int truthy = PyObject_IsTrue(marker);
            Py_DECREF(marker);
            if (truthy < 0)
                return NULL;
            if (truthy)
                continue;
        }
        /* once() wrappers: look at the original listener */
        PyObject *target = NULL;
        if (Py_TYPE(h) == &OnceWrapperType) {
            target = ((OnceWrapper *)h)->ow_listener;
            Py_INCREF(target);
        } else if (_PyObject_LookupAttr(h, s_listener, &target) < 0) {
            return NULL;
        }
        if (target != NULL && target != h) {
            if (_PyObject_LookupAttr(target, s_internal, &marker) < 0) {
                Py_DECREF(target);
                return NULL;
            }
            if (marker != NULL) {
                int truthy = PyObject_IsTrue(marker);
                Py_DECREF(marker);
                if (truthy < 0) {
                    Py_DECREF(target);
                    return NULL;
                }
                if (truthy) {
                    Py_DECREF(target);
                    continue;
                }
            }
        }
        Py_XDECREF(target);
        count++;
    }
    return PyLong_FromLong(count);
}

/* ------------------------------------------------------------------ */
/* Intrusive deque (lib/queue.js): O(1) unlink by node handle.         */
/* Ownership: the list holds ONE strong ref per linked node; q_next /  */
/* q_prev / q_queue are borrowed raw pointers, valid exactly while the */
/* node is linked (q_queue != NULL).                                   */

typedef struct NQueue NQueue;

typedef struct QNode {
    PyObject_HEAD
    PyObject *value;
    struct QNode *q_next;
    struct QNode *q_prev;
    NQueue *q_queue;
} QNode;

struct NQueue {
    PyObject_HEAD
    QNode *q_first;     /* borrowed (list ref keeps it alive) */
    QNode *q_last;      /* borrowed */
    Py_ssize_t q_len;
};

extern PyTypeObject QNodeType;
extern PyTypeObject NQueueType;

static void
nqueue_unlink(NQueue *q, QNode *n)
{
    if (n->q_prev != NULL)
        n->q_prev->q_next = n->q_next;
    else
        q->q_first = n->q_next;
    if (n->q_next != NULL)
        n->q_next->q_prev = n->q_prev;
    else
        q->q_last = n->q_prev;
    n->q_next = NULL;
    n->q_prev = NULL;
    n->q_queue = NULL;
    q->q_len--;
    Py_DECREF((PyObject *)n);   /* drop the list's ref */
}

static PyObject *
QNode_remove(PyObject *self_, PyObject *noargs)
{
    QNode *n = (QNode *)self_;
    (void)noargs;
    if (n->q_queue == NULL) {
        PyErr_SetString(PyExc_ValueError,
                        "QueueNode.remove() on unlinked node");
        return NULL;
    }
    nqueue_unlink(n->q_queue, n);
    Py_RETURN_NONE;
}

static PyObject *
QNode_get_linked(PyObject *self_, void *closure)
{
    (void)closure;
    return PyBool_FromLong(((QNode *)self_)->q_queue != NULL);
}

static PyObject *
QNode_get_value(PyObject *self_, void *closure)
{
    QNode *n = (QNode *)self_;
    (void)closure;
    if (n->value == NULL)
        Py_RETURN_NONE;
    Py_INCREF(n->value);
    return n->value;
}

static int
QNode_set_value(PyObject *self_, PyObject *v, void *closure)
{
    QNode *n = (QNode *)self_;
    (void)closure;
    Py_XINCREF(v);
    Py_XSETREF(n->value, v);
    return 0;
}

static int
QNode_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Py_VISIT(((QNode *)self_)->value);
    return 0;
}

static int
QNode_clear_(PyObject *self_)
{
    Py_CLEAR(((QNode *)self_)->value);
    return 0;
}

static void
QNode_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    QNode_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyMethodDef QNode_methods[] = {
    {"remove", QNode_remove, METH_NOARGS, NULL},
    {NULL, NULL, 0, NULL},
};

static PyGetSetDef QNode_getset[] = {
    {"linked", QNode_get_linked, NULL, NULL, NULL},
    {"value", QNode_get_value, QNode_set_value, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject QNodeType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.QueueNode",
    sizeof(QNode),
};
/* remaining slots filled in queue_types_init() — positional
 * PyTypeObject initializers are too easy to miscount */

static QNode *
nqueue_push(NQueue *q, PyObject *value)
{
    QNode *n = PyObject_GC_New(QNode, &QNodeType);
    if (n == NULL)
        return NULL;
    Py_INCREF(value);
    n->value = value;
    n->q_next = NULL;
    n->q_prev = q->q_last;
    n->q_queue = q;
    PyObject_GC_Track((PyObject *)n);
    if (q->q_last != NULL)
        q->q_last->q_next = n;
    else
        q->q_first = n;
    q->q_last = n;
    q->q_len++;
    Py_INCREF((PyObject *)n);   /* the list's ref */
    return n;                   /* caller's ref (from New) */
}

/* Pop the first value (new ref), or NULL without error if empty. */
static PyObject *
nqueue_shift_value(NQueue *q)
{
    QNode *n = q->q_first;
    if (n == NULL)
        return NULL;
    PyObject *v = n->value;
    Py_XINCREF(v);
    Py_INCREF((PyObject *)n);   /* keep alive across unlink */
    nqueue_unlink(q, n);
    Py_DECREF((PyObject *)n);
    return v ? v : Py_NewRef(Py_None);
}

static PyObject *
NQueue_push(PyObject *self_, PyObject *value)
{
    return (PyObject *)nqueue_push((NQueue *)self_, value);
}

static PyObject *
NQueue_shift(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    PyObject *v = nqueue_shift_value((NQueue *)self_);
    if (v == NULL && !PyErr_Occurred())
        PyErr_SetString(PyExc_IndexError, "shift from empty Queue");
    return v;
}

static PyObject *
NQueue_peek(PyObject *self_, PyObject *noargs)
{
    NQueue *q = (NQueue *)self_;
    (void)noargs;
    if (q->q_first == NULL) {
        PyErr_SetString(PyExc_IndexError, "peek from empty Queue");
        return NULL;
    }
    PyObject *v = q->q_first->value;
    if (v == NULL)
        Py_RETURN_NONE;
    Py_INCREF(v);
    return v;
}

static PyObject *
NQueue_is_empty(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    return PyBool_FromLong(((NQueue *)self_)->q_len == 0);
}

static PyObject *
NQueue_for_each(PyObject *self_, PyObject *cb)
{
    NQueue *q = (NQueue *)self_;
    QNode *n = q->q_first;
    Py_XINCREF((PyObject *)n);
    while (n != NULL) {
        QNode *nxt = n->q_next;
        Py_XINCREF((PyObject *)nxt);
        PyObject *args[2] = {n->value ? n->value : Py_None,
                             (PyObject *)n};
        PyObject *r = PyObject_Vectorcall(cb, args, 2, NULL);
        Py_DECREF((PyObject *)n);
        if (r == NULL) {
            Py_XDECREF((PyObject *)nxt);
            return NULL;
        }
        Py_DECREF(r);
        /* if the callback unlinked the captured successor, stop (it
         * no longer belongs to this list) — matches the pure twin's
         * generator behavior */
        if (nxt != NULL && nxt->q_queue != q) {
            Py_DECREF((PyObject *)nxt);
            break;
        }
        n = nxt;
    }
    Py_RETURN_NONE;
}

static PyObject *
NQueue_iter(PyObject *self_)
{
    /* snapshot the values; O(1)-removal users iterate rarely */
    NQueue *q = (NQueue *)self_;
    PyObject *list = PyList_New(0);
    if (list == NULL)
        return NULL;
    for (QNode *n = q->q_first; n != NULL; n = n->q_next) {
        if (PyList_Append(list, n->value ? n->value : Py_None) < 0) {
            Py_DECREF(list);
            return NULL;
        }
    }
    PyObject *it = PyObject_GetIter(list);
    Py_DECREF(list);
    return it;
}

static Py_ssize_t
NQueue_len(PyObject *self_)
{
    return ((NQueue *)self_)->q_len;
}

static PyObject *
NQueue_get_len(PyObject *self_, void *closure)
{
    (void)closure;
    return PyLong_FromSsize_t(((NQueue *)self_)->q_len);
}

static int
NQueue_traverse(PyObject *self_, visitproc visit, void *arg)
{
    NQueue *q = (NQueue *)self_;
    for (QNode *n = q->q_first; n != NULL; n = n->q_next)
        Py_VISIT((PyObject *)n);
    return 0;
}

static int
NQueue_clear_(PyObject *self_)
{
    NQueue *q = (NQueue *)self_;
    while (q->q_first != NULL) {
        QNode *n = q->q_first;
        Py_INCREF((PyObject *)n);
        nqueue_unlink(q, n);
        Py_DECREF((PyObject *)n);
    }
    return 0;
}

static void
NQueue_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    NQueue_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyObject *
NQueue_new(PyTypeObject *type, PyObject *args, PyObject *kwds)
{
    (void)args; (void)kwds;
    NQueue *q = PyObject_GC_New(NQueue, type);
    if (q == NULL)
        return NULL;
    q->q_first = NULL;
    q->q_last = NULL;
    q->q_len = 0;
    PyObject_GC_Track((PyObject *)q);
    return (PyObject *)q;
}

static PyMethodDef NQueue_methods[] = {
    {"push", NQueue_push, METH_O, NULL},
    {"shift", NQueue_shift, METH_NOARGS, NULL},
    {"peek", NQueue_peek, METH_NOARGS, NULL},
    {"is_empty", NQueue_is_empty, METH_NOARGS, NULL},
    {"for_each", NQueue_for_each, METH_O, NULL},
    {NULL, NULL, 0, NULL},
};

static PyGetSetDef NQueue_getset[] = {
    {"_len", NQueue_get_len, NULL, NULL, NULL},
    {"length", NQueue_get_len, NULL, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

static PySequenceMethods NQueue_as_sequence = {
    NQueue_len,                 /* sq_length */
};

PyTypeObject NQueueType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.Queue",
    sizeof(NQueue),
};

static void
queue_types_init(void)
{
    QNodeType.tp_dealloc = QNode_dealloc;
    QNodeType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    QNodeType.tp_traverse = QNode_traverse;
    QNodeType.tp_clear = QNode_clear_;
    QNodeType.tp_methods = QNode_methods;
    QNodeType.tp_getset = QNode_getset;

    NQueueType.tp_dealloc = NQueue_dealloc;
    NQueueType.tp_as_sequence = &NQueue_as_sequence;
    NQueueType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    NQueueType.tp_traverse = NQueue_traverse;
    NQueueType.tp_clear = NQueue_clear_;
    NQueueType.tp_iter = NQueue_iter;
    NQueueType.tp_methods = NQueue_methods;
    NQueueType.tp_getset = NQueue_getset;
    NQueueType.tp_new = NQueue_new;
}

/* ------------------------------------------------------------------ */
/* ClaimTicket: the pool's per-claim driver (pool.py _ClaimTicket) in  */
/* C.  Registered as the claim handle's stateChanged listener; on      */
/* every return to "waiting" it scans the idle queue and performs the  */
/* try_ handoff without entering Python.  Anything off the hot path    */
/* (pool not running, idle queue empty, errorOnEmpty) delegates to     */
/* pool._ticket_slow, which holds the full reference logic.            */

PyObject *s_running_st;      /* "running" */
PyObject *s_p_idleq_node;    /* "p_idleq_node" */
PyObject *s_p_total_conns;   /* "p_total_conns" */
PyObject *s_p_busy_hwm;      /* "p_busy_hwm" */
PyObject *s_p_demand_hwm;    /* "p_demand_hwm" */
PyObject *s_ticket_slow;     /* "_ticket_slow" */

typedef struct {
    PyObject_HEAD
    PyObject *ct_pool;
    PyObject *ct_handle;     /* a CHOb */
    NQueue *ct_idleq;        /* owned; NULL => always delegate */
    NQueue *ct_waiters;      /* owned */
    NQueue *ct_initq;        /* owned */
    int ct_err_on_empty;
} CTOb;

extern PyTypeObject CTType;

/* cache the pool's queues on the ticket; a pool whose queues are not
 * the native deque leaves them NULL and the ticket permanently
 * delegates to the python slow path */
static void
ct_bind_queues(CTOb *t, PyObject *pool)
{
    PyObject *iq = PyObject_GetAttrString(pool, "p_idleq");
    PyObject *wq = iq ? PyObject_GetAttrString(pool, "p_waiters") : NULL;
    PyObject *nq = wq ? PyObject_GetAttrString(pool, "p_initq") : NULL;
    if (nq == NULL) {
        PyErr_Clear();
        Py_XDECREF(iq);
        Py_XDECREF(wq);
        return;
    }
    if (PyObject_TypeCheck(iq, &NQueueType) &&
        PyObject_TypeCheck(wq, &NQueueType) &&
        PyObject_TypeCheck(nq, &NQueueType)) {
        t->ct_idleq = (NQueue *)iq;
        t->ct_waiters = (NQueue *)wq;
        t->ct_initq = (NQueue *)nq;
    } else {
        Py_DECREF(iq);
        Py_DECREF(wq);
        Py_DECREF(nq);
    }
}

static int
ct_note_demand(CTOb *t)
{
    /* O(1) inline of pool._note_demand() */
    Py_ssize_t nw = t->ct_waiters->q_len;
    Py_ssize_t ni = t->ct_initq->q_len;
    Py_ssize_t spares = t->ct_idleq->q_len + ni - nw;
    if (spares < 0)
        spares = 0;
    PyObject *tot = PyObject_GetAttr(t->ct_pool, s_p_total_conns);
    if (tot == NULL)
        return -1;
    long total = PyLong_AsLong(tot);
    Py_DECREF(tot);
    if (total == -1 && PyErr_Occurred())
        return -1;
    long busy = total - (long)spares;
    if (busy < 0)
        busy = 0;
    long extras = (long)(nw - ni);
    if (extras < 0)
        extras = 0;

    PyObject *cur = PyObject_GetAttr(t->ct_pool, s_p_busy_hwm);
    if (cur == NULL)
        return -1;
    long curv = PyLong_AsLong(cur);
    Py_DECREF(cur);
    if (curv == -1 && PyErr_Occurred())
        return -1;
    if (busy > curv) {
        PyObject *nv = PyLong_FromLong(busy);
        if (nv == NULL)
            return -1;
        int r = PyObject_SetAttr(t->ct_pool, s_p_busy_hwm, nv);
        Py_DECREF(nv);
        if (r < 0)
            return -1;
    }
    cur = PyObject_GetAttr(t->ct_pool, s_p_demand_hwm);
    if (cur == NULL)
        return -1;
    curv = PyLong_AsLong(cur);
    Py_DECREF(cur);
    if (curv == -1 && PyErr_Occurred())
        return -1;
    if (busy + extras > curv) {
        PyObject *nv = PyLong_FromLong(busy + extras);
        if (nv == NULL)
            return -1;
        int r = PyObject_SetAttr(t->ct_pool, s_p_demand_hwm, nv);
        Py_DECREF(nv);
        if (r < 0)
            return -1;
    }
    return 0;
}

PyObject *s_p_counters;          /* "p_counters" */
PyObject *s_ch_waiter_node;      /* "ch_waiter_node" */
PyObject *s_p_rebal_scheduled;   /* "p_rebal_scheduled" */
PyObject *s_max_claim_queue;     /* "max-claim-queue" */
PyObject *s_queued_claim;        /* "queued-claim" */
extern PyObject *s_rebalance;    /* "rebalance" (SlotDispatch section) */

/* the waiter-enqueue branch of pool._ticket_slow in C: push the
 * handle, fold the demand sample, bump the HWM/queued counters, and
 * schedule a rebalance unless one is already pending */
static int
ct_enqueue_waiter(CTOb *t)
{
    QNode *n = nqueue_push(t->ct_waiters, t->ct_handle);
    if (n == NULL)
        return -1;
    /* the handle keeps its node so leaving 'waiting' (timeout/
     * cancel) unlinks it immediately — overload cannot accumulate
     * dead queue entries between feeds (see pool._ticket_slow) */
    int rc = PyObject_SetAttr(t->ct_handle, s_ch_waiter_node,
                              (PyObject *)n);
    Py_DECREF((PyObject *)n);
    if (rc < 0)
        return -1;
    if (ct_note_demand(t) < 0)
        return -1;

    PyObject *counters = PyObject_GetAttr(t->ct_pool, s_p_counters);
    if (counters == NULL)
        return -1;
    if (!PyDict_Check(counters)) {
        Py_DECREF(counters);
        PyErr_SetString(PyExc_TypeError, "p_counters must be a dict");
        return -1;
    }
    /* max-claim-queue HWM */
    PyObject *cur = PyDict_GetItemWithError(counters, s_max_claim_queue);
    if (cur == NULL && PyErr_Occurred()) {
        Py_DECREF(counters);
        return -1;
    }
    long curv = -1;
    if (cur != NULL) {
        curv = PyLong_AsLong(cur);
        if (curv == -1 && PyErr_Occurred()) {
            Py_DECREF(counters);
            return -1;
        }
    }
    if ((long)t->ct_waiters->q_len > curv) {
        PyObject *nv = PyLong_FromSsize_t(t->ct_waiters->q_len);
        if (nv == NULL || PyDict_SetItem(counters, s_max_claim_queue,
                                         nv) < 0) {
            Py_XDECREF(nv);
            Py_DECREF(counters);
            return -1;
        }
        Py_DECREF(nv);
    }
    /* queued-claim += 1 (not a tracked metric event) */
    cur = PyDict_GetItemWithError(counters, s_queued_claim);
    if (cur == NULL && PyErr_Occurred()) {
        Py_DECREF(counters);
        return -1;
    }
    long qc = 0;
    if (cur != NULL) {
        qc = PyLong_AsLong(cur);
        if (qc == -1 && PyErr_Occurred()) {
            Py_DECREF(counters);
            return -1;
        }
    }
    PyObject *nv = PyLong_FromLong(qc + 1);
    if (nv == NULL ||
        PyDict_SetItem(counters, s_queued_claim, nv) < 0) {
        Py_XDECREF(nv);
        Py_DECREF(counters);
        return -1;
    }
    Py_DECREF(nv);
    Py_DECREF(counters);

    /* rebalance() unless one is already scheduled (the common case
     * under sustained queueing) */
    PyObject *sched = PyObject_GetAttr(t->ct_pool, s_p_rebal_scheduled);
    if (sched == NULL)
        return -1;
    int pending = PyObject_IsTrue(sched);
    Py_DECREF(sched);
    if (pending < 0)
        return -1;
    if (!pending) {
        PyObject *r = PyObject_CallMethodObjArgs(t->ct_pool,
                                                 s_rebalance, NULL);
        if (r == NULL)
            return -1;
        Py_DECREF(r);
    }
    return 0;
}

static int
ct_slow(CTOb *t)
{
    PyObject *r = PyObject_CallMethodObjArgs(
        t->ct_pool, s_ticket_slow, t->ct_handle,
        t->ct_err_on_empty ? Py_True : Py_False, NULL);
    if (r == NULL)
        return -1;
    Py_DECREF(r);
    return 0;
}

static int
ct_try_next(CTOb *t)
{
    CHOb *h = (CHOb *)t->ct_handle;
    if (!ch_state_is(h, s_waiting))
        return 0;
    if (t->ct_idleq == NULL)
        return ct_slow(t);
    PyObject *pst = ((FSMOb *)t->ct_pool)->f_state;
    if (!(pst == s_running_st ||
          (pst != NULL && PyUnicode_Compare(pst, s_running_st) == 0)))
        return ct_slow(t);

    NQueue *q = t->ct_idleq;
    while (q->q_len > 0) {
        QNode *n = q->q_first;
        PyObject *fsm = n->value;
        Py_XINCREF(fsm);
        Py_INCREF((PyObject *)n);
        nqueue_unlink(q, n);
        Py_DECREF((PyObject *)n);
        if (fsm == NULL)
            continue;
        if (PyObject_SetAttr(fsm, s_p_idleq_node, Py_None) < 0) {
            Py_DECREF(fsm);
            return -1;
        }
        /* stale entries tolerated: only a slot still in 'idle' is
         * usable (lib/pool.js:934-951) */
        int isidle;
        if (PyObject_TypeCheck(fsm, &FSMType)) {
            PyObject *fs = ((FSMOb *)fsm)->f_state;
            isidle = (fs == s_idle ||
                      (fs != NULL &&
                       PyUnicode_Compare(fs, s_idle) == 0));
        } else {
            PyObject *r = PyObject_CallMethodObjArgs(
                fsm, s_is_in_state, s_idle, NULL);
            if (r == NULL) {
                Py_DECREF(fsm);
                return -1;
            }
            isidle = PyObject_IsTrue(r);
            Py_DECREF(r);
        }
        if (!isidle) {
            Py_DECREF(fsm);
            continue;
        }
        PyObject *r = CH_try_((PyObject *)h, fsm);
        Py_DECREF(fsm);
        if (r == NULL)
            return -1;
        Py_DECREF(r);
        return ct_note_demand(t);
    }
    /* idle queue empty: queue as a waiter (C) unless errorOnEmpty
     * semantics apply, which keep the full python logic */
    if (!t->ct_err_on_empty)
        return ct_enqueue_waiter(t);
    return ct_slow(t);
}

static PyObject *
CT_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    CTOb *t = (CTOb *)self_;
    (void)kwds;
    PyObject *st;
    if (PyTuple_GET_SIZE(args) != 1) {
        PyErr_SetString(PyExc_TypeError, "ClaimTicket(state)");
        return NULL;
    }
    st = PyTuple_GET_ITEM(args, 0);
    if (st == s_waiting ||
        (PyUnicode_Check(st) && PyUnicode_Compare(st, s_waiting) == 0)) {
        if (ct_try_next(t) < 0)
            return NULL;
    }
    Py_RETURN_NONE;
}

static int
CT_traverse(PyObject *self_, visitproc visit, void *arg)
{
    CTOb *t = (CTOb *)self_;
    Py_VISIT(t->ct_pool);
    Py_VISIT(t->ct_handle);
    Py_VISIT((PyObject *)t->ct_idleq);
    Py_VISIT((PyObject *)t->ct_waiters);
    Py_VISIT((PyObject *)t->ct_initq);
    return 0;
}

static int
CT_clear_(PyObject *self_)
{
    CTOb *t = (CTOb *)self_;
    Py_CLEAR(t->ct_pool);
    Py_CLEAR(t->ct_handle);
    Py_CLEAR(t->ct_idleq);
    Py_CLEAR(t->ct_waiters);
    Py_CLEAR(t->ct_initq);
    return 0;
}

static void
CT_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    CT_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyObject *
CT_new(PyTypeObject *type, PyObject *args, PyObject *kwds)
{
    PyObject *pool, *handle, *err_on_empty;
    (void)kwds;
    if (!PyArg_ParseTuple(args, "OOO", &pool, &handle, &err_on_empty))
        return NULL;
    if (!PyObject_TypeCheck(pool, &FSMType)) {
        PyErr_SetString(PyExc_TypeError, "pool must be a native FSM");
        return NULL;
    }
    if (!PyObject_TypeCheck(handle, &CHType)) {
        PyErr_SetString(PyExc_TypeError,
                        "handle must be a ClaimHandleBase");
        return NULL;
    }
    CTOb *t = PyObject_GC_New(CTOb, type);
    if (t == NULL)
        return NULL;
    Py_INCREF(pool);
    t->ct_pool = pool;
    Py_INCREF(handle);
    t->ct_handle = handle;
    t->ct_idleq = NULL;
    t->ct_waiters = NULL;
    t->ct_initq = NULL;
    t->ct_err_on_empty = PyObject_IsTrue(err_on_empty);
    PyObject_GC_Track((PyObject *)t);
    ct_bind_queues(t, pool);
    return (PyObject *)t;
}

PyTypeObject CTType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.ClaimTicket",
    sizeof(CTOb),
};

/* pool_claim(handle_cls, pool, cb, log, loop) -> handle | True
 * The whole no-options/no-CoDel pool.claim() body in C: bump the
 * claim counter, check the pool state (returns True when
 * stopping/stopped/failed so python builds the short-circuit error
 * without re-counting), capture the claim stack, and build
 * handle+ticket via the claim_fast core with timeout=inf. */
static PyObject *speed_claim_fast(PyObject *mod, PyObject *const *args,
                                  Py_ssize_t nargs);

static PyObject *
speed_pool_claim(PyObject *mod, PyObject *const *args, Py_ssize_t nargs)
{
    (void)mod;
    if (nargs != 5) {
        PyErr_SetString(PyExc_TypeError,
                        "pool_claim(cls, pool, cb, log, loop)");
        return NULL;
    }
    PyObject *pool = args[1];
    if (!PyObject_TypeCheck(pool, &FSMType)) {
        PyErr_SetString(PyExc_TypeError, "pool must be a native FSM");
        return NULL;
    }
    /* counters["claim"] += 1 ("claim" is not a tracked metric event) */
    PyObject *counters = PyObject_GetAttr(pool, s_p_counters);
    if (counters == NULL)
        return NULL;
    if (PyDict_Check(counters)) {
        PyObject *cur = PyDict_GetItemWithError(counters, s_claim);
        if (cur == NULL && PyErr_Occurred()) {
            Py_DECREF(counters);
            return NULL;
        }
        long v = 0;
        if (cur != NULL) {
            v = PyLong_AsLong(cur);
            if (v == -1 && PyErr_Occurred()) {
                Py_DECREF(counters);
                return NULL;
            }
        }
        PyObject *nv = PyLong_FromLong(v + 1);
        if (nv == NULL ||
            PyDict_SetItem(counters, s_claim, nv) < 0) {
            Py_XDECREF(nv);
            Py_DECREF(counters);
            return NULL;
        }
        Py_DECREF(nv);
    }
    Py_DECREF(counters);

    PyObject *pst = ((FSMOb *)pool)->f_state;
    int bad = 0;
    if (pst == NULL) {
        bad = 1;
    } else if (!(pst == s_running_st ||
                 PyUnicode_Compare(pst, s_running_st) == 0)) {
        /* only starting is also claim-able */
        bad = !(PyUnicode_CompareWithASCIIString(pst, "starting") == 0);
    }
    if (bad)
        Py_RETURN_TRUE;

    PyObject *stack = PyObject_CallNoArgs(g_capture_stack);
    if (stack == NULL)
        return NULL;
    PyObject *inf = PyFloat_FromDouble(HUGE_VAL);
    if (inf == NULL) {
        Py_DECREF(stack);
        return NULL;
    }
    PyObject *cfargs[8] = {args[0], pool, stack, args[2], args[3],
                           inf, args[4], Py_False};
    PyObject *h = speed_claim_fast(NULL, cfargs, 8);
    Py_DECREF(stack);
    Py_DECREF(inf);
    return h;
}

/* claim_fast(handle_cls, pool, stack, cb, log, timeout, loop,
 *            err_on_empty) -> handle
 * One C call for pool.claim's hot tail: allocate the ClaimHandle
 * subclass, run the _setup core (throw_error=True), build the native
 * ticket and register it as the stateChanged listener. */
PyObject *CH__setup(PyObject *self_, PyObject *const *args,
                    Py_ssize_t nargs);

static PyObject *
speed_claim_fast(PyObject *mod, PyObject *const *args, Py_ssize_t nargs)
{
    (void)mod;
    if (nargs != 8) {
        PyErr_SetString(PyExc_TypeError,
            "claim_fast(cls, pool, stack, cb, log, timeout, loop, "
            "err_on_empty)");
        return NULL;
    }
    PyTypeObject *cls = (PyTypeObject *)args[0];
    if (!PyType_Check(args[0]) ||
        !PyType_IsSubtype(cls, &CHType)) {
        PyErr_SetString(PyExc_TypeError,
                        "cls must be a ClaimHandleBase subclass");
        return NULL;
    }
    PyObject *pool = args[1];
    if (!PyObject_TypeCheck(pool, &FSMType)) {
        PyErr_SetString(PyExc_TypeError, "pool must be a native FSM");
        return NULL;
    }
    int err_on_empty = PyObject_IsTrue(args[7]);
    if (err_on_empty < 0)
        return NULL;

    PyObject *handle = cls->tp_alloc(cls, 0);
    if (handle == NULL)
        return NULL;
    PyObject *setup_args[7] = {pool, args[2], args[3], args[4],
                               args[5], Py_True, args[6]};
    PyObject *r = CH__setup(handle, setup_args, 7);
    if (r == NULL) {
        Py_DECREF(handle);
        return NULL;
    }
    Py_DECREF(r);

    CTOb *t = PyObject_GC_New(CTOb, &CTType);
    if (t == NULL) {
        Py_DECREF(handle);
        return NULL;
    }
    Py_INCREF(pool);
    t->ct_pool = pool;
    Py_INCREF(handle);
    t->ct_handle = handle;
    t->ct_idleq = NULL;
    t->ct_waiters = NULL;
    t->ct_initq = NULL;
    t->ct_err_on_empty = err_on_empty;
    PyObject_GC_Track((PyObject *)t);
    ct_bind_queues(t, pool);
    r = emitter_add((Emitter *)handle, s_stateChanged, (PyObject *)t);
    Py_DECREF((PyObject *)t);
    if (r == NULL) {
        Py_DECREF(handle);
        return NULL;
    }
    Py_DECREF(r);
    return handle;
}

static void
ct_type_init(void)
{
    CTType.tp_dealloc = CT_dealloc;
    CTType.tp_call = CT_call;
    CTType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    CTType.tp_traverse = CT_traverse;
    CTType.tp_clear = CT_clear_;
    CTType.tp_new = CT_new;
}

/* Terminal-state cycle breaker (see fsm_enter_loop): dispose the
 * terminal scope (fsm<->scope cycle) and unregister the pool ticket
 * from the handle's stateChanged listeners (handle<->ticket cycle).
 * Both are semantically invisible — the terminal scope registers
 * nothing, and the ticket only reacts to 'waiting', which is
 * unreachable from a terminal state.  User listeners are untouched
 * and still receive the queued terminal stateChanged. */
void
ch_terminal_cleanup(FSMOb *self)
{
    /* unlink from the waiter queue if still queued (timeout/cancel
     * while waiting): keeps the queue free of dead entries */
    PyObject *wn = PyObject_GetAttr((PyObject *)self, s_ch_waiter_node);
    if (wn == NULL) {
        PyErr_Clear();
    } else {
        if (PyObject_TypeCheck(wn, &QNodeType) &&
            ((QNode *)wn)->q_queue != NULL)
            nqueue_unlink(((QNode *)wn)->q_queue, (QNode *)wn);
        Py_DECREF(wn);
        if (PyObject_SetAttr((PyObject *)self, s_ch_waiter_node,
                             Py_None) < 0)
            PyErr_Clear();
    }
    if (self->f_scope != NULL) {
        PyObject *old = self->f_scope;
        self->f_scope = NULL;
        PyObject *r = Scope_dispose(old, NULL);
        Py_DECREF(old);
        if (r == NULL)
            PyErr_Clear();
        else
            Py_DECREF(r);
    }
    Emitter *em = &self->base;
    if (em->ev_events == NULL)
        return;
    PyObject *ls = PyDict_GetItemWithError(em->ev_events, s_stateChanged);
    if (ls == NULL) {
        PyErr_Clear();
        return;
    }
    if (!PyList_Check(ls))
        return;
    for (Py_ssize_t i = PyList_GET_SIZE(ls) - 1; i >= 0; i--) {
        if (PyObject_TypeCheck(PyList_GET_ITEM(ls, i), &CTType)) {
            if (PyList_SetSlice(ls, i, i + 1, NULL) < 0) {
                PyErr_Clear();
                return;
            }
        }
    }
}

/* ------------------------------------------------------------------ */
/* SlotKit: native entries for ConnectionSlotFSM's busy/idle hot       */
/* cycle (connection_fsm.py state_busy/state_idle).  One claim/release */
/* runs idle -> busy -> idle; with the kit installed both entries and  */
/* their event callbacks execute in C, with zero per-cycle Python      */
/* allocation.  Installed only for slots without a health checker;     */
/* monitor/unwanted idle entries fall back to the Python entry, and    */
/* all exit transitions out of the cycle enter Python states.          */

PyObject *s_connected_st;    /* "connected" */
PyObject *s_busy_st;         /* "busy" */
PyObject *s_stopping_st;     /* "stopping" */
PyObject *s_stopped_st;      /* "stopped" */
PyObject *s_retrying_st;     /* "retrying" */
PyObject *s_connecting_st;   /* "connecting" */
PyObject *s_killing_st;      /* "killing" */
PyObject *s_unwanted_evt;    /* "unwanted" */
PyObject *s_csf_handle;      /* "csf_handle" */
PyObject *s_csf_prev_handle; /* "csf_prev_handle" */
PyObject *s_csf_wanted;      /* "csf_wanted" */
PyObject *s_csf_monitor;     /* "csf_monitor" */
PyObject *s_sm_socket;       /* "sm_socket" */
PyObject *t_busy_valid;      /* like _BUSY_VALID */
PyObject *t_idle_valid;      /* like _IDLE_VALID */

typedef struct SlotKit SlotKit;

/* preallocated listener callables; `which` selects the behavior */
enum { KCB_BUSY_SMGR = 0, KCB_BUSY_HDL, KCB_IDLE_SMGR, KCB_IDLE_UNWANTED };

typedef struct {
    PyObject_HEAD
    SlotKit *kc_kit;            /* owned */
    int kc_which;
} KitCb;

struct SlotKit {
    PyObject_HEAD
    FSMOb *k_slot;              /* owned */
    PyObject *k_smgr;           /* owned; an FSMOb */
    PyObject *k_observed;       /* event-observed smgr state, owned */
    Scope *k_busy_scope;        /* owned; scope of the last busy entry */
    Scope *k_idle_scope;        /* owned; scope of the last idle entry */
    PyObject *k_cbs[4];         /* owned KitCb instances */
};

extern PyTypeObject SlotKitType;
extern PyTypeObject KitCbType;

static int
kit_truthy_attr(PyObject *obj, PyObject *name)
{
    PyObject *v = PyObject_GetAttr(obj, name);
    if (v == NULL)
        return -1;
    int t = PyObject_IsTrue(v);
    Py_DECREF(v);
    return t;
}

static int
kit_scope_goto(Scope *scope, PyObject *state)
{
    if (scope == NULL || !scope->sc_active)
        return 0;  /* stale handler in the same cascade: obsolete */
    return fsm_goto_state((FSMOb *)scope->sc_fsm, state);
}

static int
kit_busy_on_release(SlotKit *k)
{
    Scope *S = k->k_busy_scope;
    PyObject *st = k->k_observed;
    int wanted = kit_truthy_attr((PyObject *)k->k_slot, s_csf_wanted);
    if (wanted < 0)
        return -1;
    if (st == s_connected_st ||
        (st != NULL && PyUnicode_Compare(st, s_connected_st) == 0))
        return kit_scope_goto(S, wanted ? s_idle : s_stopping_st);
    if (st == s_closed || PyUnicode_Compare(st, s_closed) == 0)
        return kit_scope_goto(S, wanted ? s_connecting_st : s_stopped_st);
    if (st == s_error_evt || PyUnicode_Compare(st, s_error_evt) == 0)
        return kit_scope_goto(S, s_retrying_st);
    PyErr_Format(g_fsm_error,
                 "Handle released while smgr was in unhandled state %R",
                 st ? st : Py_None);
    return -1;
}

static int
kit_busy_on_close(SlotKit *k)
{
    Scope *S = k->k_busy_scope;
    PyObject *st = k->k_observed;
    if (st == s_connected_st ||
        (st != NULL && PyUnicode_Compare(st, s_connected_st) == 0))
        return kit_scope_goto(S, s_killing_st);
    return kit_scope_goto(S, s_retrying_st);
}

static int
kit_idle_smgr_changed(SlotKit *k, PyObject *st)
{
    Scope *S = k->k_idle_scope;
    int wanted = kit_truthy_attr((PyObject *)k->k_slot, s_csf_wanted);
    if (wanted < 0)
        return -1;
    if (st == s_error_evt || PyUnicode_Compare(st, s_error_evt) == 0)
        return kit_scope_goto(S, wanted ? s_retrying_st : s_stopped_st);
    if (st == s_closed || PyUnicode_Compare(st, s_closed) == 0)
        return kit_scope_goto(S, wanted ? s_connecting_st : s_stopped_st);
    PyErr_Format(g_fsm_error,
                 "Unhandled smgr state transition: connected => %R", st);
    return -1;
}

static int
kit_idle_unwanted(SlotKit *k)
{
    /* mirror of _idle_on_unwanted incl. the same-spin death fix */
    Scope *S = k->k_idle_scope;
    PyObject *sst = ((FSMOb *)k->k_smgr)->f_state;
    if (sst == s_connected_st ||
        (sst != NULL && PyUnicode_Compare(sst, s_connected_st) == 0))
        return kit_scope_goto(S, s_stopping_st);
    if (sst == s_error_evt || sst == s_closed ||
        (sst != NULL && (PyUnicode_Compare(sst, s_error_evt) == 0 ||
                         PyUnicode_Compare(sst, s_closed) == 0)))
        return kit_scope_goto(S, s_stopped_st);
    return 0;
}

static PyObject *
KitCb_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    KitCb *cb = (KitCb *)self_;
    SlotKit *k = cb->kc_kit;
    PyObject *st = NULL;
    (void)kwds;
    if (PyTuple_GET_SIZE(args) >= 1)
        st = PyTuple_GET_ITEM(args, 0);
    int r = 0;
    switch (cb->kc_which) {
    case KCB_BUSY_SMGR:
        /* record the smgr state as observed through events
         * (lib/connection-fsm.js:885-890) */
        if (st != NULL) {
            Py_INCREF(st);
            Py_XSETREF(k->k_observed, st);
        }
        break;
    case KCB_BUSY_HDL:
        if (st == s_released ||
            (st != NULL && PyUnicode_Compare(st, s_released) == 0))
            r = kit_busy_on_release(k);
        else if (st == s_closed ||
                 (st != NULL && PyUnicode_Compare(st, s_closed) == 0))
            r = kit_busy_on_close(k);
        break;
    case KCB_IDLE_SMGR:
        r = kit_idle_smgr_changed(k, st);
        break;
    case KCB_IDLE_UNWANTED:
        r = kit_idle_unwanted(k);
        break;
    }
    if (r < 0)
        return NULL;
    Py_RETURN_NONE;
}

static int
KitCb_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Py_VISIT((PyObject *)((KitCb *)self_)->kc_kit);
    return 0;
}

static int
KitCb_clear_(PyObject *self_)
{
    Py_CLEAR(((KitCb *)self_)->kc_kit);
    return 0;
}

static void
KitCb_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    KitCb_clear_(self_);
    PyObject_GC_Del(self_);
}

PyTypeObject KitCbType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed._SlotKitCb",
    sizeof(KitCb),
};

/* -- the native state entries --------------------------------------- */

static int
kit_state_busy(SlotKit *k, Scope *scope)
{
    FSMOb *slot = k->k_slot;
    Py_INCREF(t_busy_valid);
    Py_XSETREF(slot->f_valid, t_busy_valid);

    PyObject *hdl = PyObject_GetAttr((PyObject *)slot, s_csf_handle);
    if (hdl == NULL)
        return -1;
    Py_INCREF(s_connected_st);
    Py_XSETREF(k->k_observed, s_connected_st);
    Py_INCREF((PyObject *)scope);
    Py_XSETREF(k->k_busy_scope, scope);

    if (scope_on_c(scope, k->k_smgr, s_stateChanged,
                   k->k_cbs[KCB_BUSY_SMGR]) < 0 ||
        scope_on_c(scope, hdl, s_stateChanged,
                   k->k_cbs[KCB_BUSY_HDL]) < 0) {
        Py_DECREF(hdl);
        return -1;
    }

    PyObject *sst = ((FSMOb *)k->k_smgr)->f_state;
    int connected = (sst == s_connected_st ||
                     (sst != NULL &&
                      PyUnicode_Compare(sst, s_connected_st) == 0));
    PyObject *r;
    if (connected) {
        PyObject *sock = PyObject_GetAttr(k->k_smgr, s_sm_socket);
        if (sock == NULL) {
            Py_DECREF(hdl);
            return -1;
        }
        r = CH_accept(hdl, sock);
        Py_DECREF(sock);
        Py_DECREF(hdl);
        if (r == NULL)
            return -1;
        Py_DECREF(r);
    } else {
        /* lost the race with the smgr: treat as released
         * (lib/connection-fsm.js:1129-1196) */
        r = CH_reject(hdl, NULL);
        Py_DECREF(hdl);
        if (r == NULL)
            return -1;
        Py_DECREF(r);
        if (PyObject_SetAttr((PyObject *)slot, s_csf_handle,
                             Py_None) < 0)
            return -1;
        if (kit_busy_on_release(k) < 0)
            return -1;
    }
    return 1;
}

static int
kit_state_idle(SlotKit *k, Scope *scope)
{
    FSMOb *slot = k->k_slot;
    /* the monitor-conversion and unwanted cases keep the full Python
     * entry (cold paths with documented divergences) */
    int monitor = kit_truthy_attr((PyObject *)slot, s_csf_monitor);
    if (monitor < 0)
        return -1;
    if (monitor)
        return 0;
    int wanted = kit_truthy_attr((PyObject *)slot, s_csf_wanted);
    if (wanted < 0)
        return -1;
    if (!wanted)
        return 0;

    Py_INCREF(t_idle_valid);
    Py_XSETREF(slot->f_valid, t_idle_valid);
    Py_INCREF((PyObject *)scope);
    Py_XSETREF(k->k_idle_scope, scope);

    PyObject *hdl = PyObject_GetAttr((PyObject *)slot, s_csf_handle);
    if (hdl == NULL)
        return -1;
    if (hdl != Py_None) {
        if (PyObject_SetAttr((PyObject *)slot, s_csf_prev_handle,
                             hdl) < 0) {
            Py_DECREF(hdl);
            return -1;
        }
        if (PyObject_SetAttr((PyObject *)slot, s_csf_handle,
                             Py_None) < 0) {
            Py_DECREF(hdl);
            return -1;
        }
    }
    Py_DECREF(hdl);

    if (scope_on_c(scope, k->k_smgr, s_stateChanged,
                   k->k_cbs[KCB_IDLE_SMGR]) < 0)
        return -1;
    if (scope_on_c(scope, (PyObject *)slot, s_unwanted_evt,
                   k->k_cbs[KCB_IDLE_UNWANTED]) < 0)
        return -1;
    return 1;
}

int
slotkit_maybe_entry(PyObject *kit_, FSMOb *fsm, PyObject *target,
                    Scope *scope)
{
    SlotKit *k = (SlotKit *)kit_;
    if (k->k_slot != fsm)
        return 0;
    if (target == s_busy_st ||
        (PyUnicode_Check(target) &&
         PyUnicode_Compare(target, s_busy_st) == 0))
        return kit_state_busy(k, scope);
    if (target == s_idle ||
        (PyUnicode_Check(target) &&
         PyUnicode_Compare(target, s_idle) == 0))
        return kit_state_idle(k, scope);
    return 0;
}

/* ------------------------------------------------------------------ */
/* ControlledDelay: the CoDel AQM (codel.py / lib/codel.js) in C so    */
/* the overload-shed feed path stays native.                           */

#define CODEL_INTERVAL_MS 100.0

typedef struct {
    PyObject_HEAD
    PyObject *cd_loop;          /* owned */
    double cd_targdelay;
    double cd_first_above_time;
    double cd_drop_next;
    long cd_count;
    int cd_dropping;
    double cd_last_empty;
} CoDelOb;

extern PyTypeObject CoDelType;

static double
codel_now(CoDelOb *self, int *err)
{
    PyObject *t = PyObject_CallMethodObjArgs(self->cd_loop, s_time, NULL);
    if (t == NULL) {
        *err = 1;
        return 0.0;
    }
    double secs = PyFloat_AsDouble(t);
    Py_DECREF(t);
    if (secs == -1.0 && PyErr_Occurred()) {
        *err = 1;
        return 0.0;
    }
    *err = 0;
    return secs * 1000.0;
}

static int
codel_can_drop(CoDelOb *self, double now, double start)
{
    double sojourn = now - start;
    if (sojourn < self->cd_targdelay)
        self->cd_first_above_time = 0.0;
    else if (self->cd_first_above_time == 0.0)
        self->cd_first_above_time = now + CODEL_INTERVAL_MS;
    else if (now >= self->cd_first_above_time)
        return 1;
    return 0;
}

/* int result: 1 drop, 0 keep, -1 error.  Mirrors
 * ControlledDelay.overloaded() exactly. */
static int
codel_overloaded_c(CoDelOb *self, double start)
{
    int err = 0;
    double now = codel_now(self, &err);
    if (err)
        return -1;
    int ok_to_drop = codel_can_drop(self, now, start);
    int drop_claim = 0;
    if (self->cd_dropping) {
        if (!ok_to_drop) {
            self->cd_dropping = 0;
        } else if (now >= self->cd_drop_next) {
            /* NOTE: the reference deliberately does NOT advance
             * cd_drop_next here (lib/codel.js:62-68): once dropping
             * and past drop-next, every dequeue above target drops
             * until the sojourn falls below target again. */
            drop_claim = 1;
            self->cd_count++;
        }
    } else if (ok_to_drop &&
               ((now - self->cd_drop_next < CODEL_INTERVAL_MS) ||
                (now - self->cd_first_above_time >=
                 CODEL_INTERVAL_MS))) {
        drop_claim = 1;
        self->cd_dropping = 1;
        if (now - self->cd_drop_next < CODEL_INTERVAL_MS)
            self->cd_count = self->cd_count > 2 ?
                self->cd_count - 2 : 1;
        else
            self->cd_count = 1;
        self->cd_drop_next = now +
            CODEL_INTERVAL_MS / sqrt((double)self->cd_count);
    }
    return drop_claim;
}

static PyObject *
CoDel_overloaded(PyObject *self_, PyObject *start_)
{
    double start = PyFloat_AsDouble(start_);
    if (start == -1.0 && PyErr_Occurred())
        return NULL;
    int r = codel_overloaded_c((CoDelOb *)self_, start);
    if (r < 0)
        return NULL;
    return PyBool_FromLong(r);
}

static PyObject *
CoDel_can_drop(PyObject *self_, PyObject *const *args, Py_ssize_t nargs)
{
    if (nargs != 2) {
        PyErr_SetString(PyExc_TypeError, "can_drop(now, start)");
        return NULL;
    }
    double now = PyFloat_AsDouble(args[0]);
    if (now == -1.0 && PyErr_Occurred())
        return NULL;
    double start = PyFloat_AsDouble(args[1]);
    if (start == -1.0 && PyErr_Occurred())
        return NULL;
    return PyBool_FromLong(codel_can_drop((CoDelOb *)self_, now, start));
}

static PyObject *
CoDel_get_drop_next(PyObject *self_, PyObject *now_)
{
    CoDelOb *self = (CoDelOb *)self_;
    double now = PyFloat_AsDouble(now_);
    if (now == -1.0 && PyErr_Occurred())
        return NULL;
    return PyFloat_FromDouble(
        now + CODEL_INTERVAL_MS / sqrt((double)self->cd_count));
}

static int
codel_empty_c(CoDelOb *self)
{
    int err = 0;
    double now = codel_now(self, &err);
    if (err)
        return -1;
    self->cd_last_empty = now;
    self->cd_first_above_time = 0.0;
    return 0;
}

static PyObject *
CoDel_empty(PyObject *self_, PyObject *noargs)
{
    (void)noargs;
    if (codel_empty_c((CoDelOb *)self_) < 0)
        return NULL;
    Py_RETURN_NONE;
}

static PyObject *
CoDel_get_max_idle(PyObject *self_, PyObject *noargs)
{
    CoDelOb *self = (CoDelOb *)self_;
    (void)noargs;
    double bound = self->cd_targdelay * 10.0;
    int err = 0;
    double now = codel_now(self, &err);
    if (err)
        return NULL;
    if (self->cd_last_empty < now - bound)
        return PyFloat_FromDouble(self->cd_targdelay * 3.0);
    return PyFloat_FromDouble(bound);
}

static PyObject *
CoDel_get_dropping(PyObject *self_, void *closure)
{
    (void)closure;
    return PyBool_FromLong(((CoDelOb *)self_)->cd_dropping);
}

static int
CoDel_traverse(PyObject *self_, visitproc visit, void *arg)
{
    Py_VISIT(((CoDelOb *)self_)->cd_loop);
    return 0;
}

static int
CoDel_clear_(PyObject *self_)
{
    Py_CLEAR(((CoDelOb *)self_)->cd_loop);
    return 0;
}

static void
CoDel_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    CoDel_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyObject *
CoDel_new(PyTypeObject *type, PyObject *args, PyObject *kwds)
{
    PyObject *target, *loop = Py_None;
    static const char *kwlist[] = {"target_claim_delay", "loop", NULL};
    if (!PyArg_ParseTupleAndKeywords(args, kwds, "O|O",
                                     const_cast<char **>(kwlist),
                                     &target, &loop))
        return NULL;
    double targ;
    if (PyFloat_Check(target) || PyLong_Check(target)) {
        targ = PyFloat_AsDouble(target);
        if (targ == -1.0 && PyErr_Occurred())
            return NULL;
    } else {
        PyErr_SetString(PyExc_ValueError,
                        "target_claim_delay must be finite");
        return NULL;
    }
    if (!std::isfinite(targ)) {
        PyErr_SetString(PyExc_ValueError,
                        "target_claim_delay must be finite");
        return NULL;
    }
    PyObject *resolved = PyObject_CallOneArg(g_get_loop, loop);
    if (resolved == NULL)
        return NULL;
    CoDelOb *self = PyObject_GC_New(CoDelOb, type);
    if (self == NULL) {
        Py_DECREF(resolved);
        return NULL;
    }
    self->cd_loop = resolved;
    self->cd_targdelay = targ;
    self->cd_first_above_time = 0.0;
    self->cd_drop_next = 0.0;
    self->cd_count = 0;
    self->cd_dropping = 0;
    self->cd_last_empty = 0.0;
    PyObject_GC_Track((PyObject *)self);
    /* treat the queue as having just been empty (see codel.py) */
    int err = 0;
    double now = codel_now(self, &err);
    if (err) {
        Py_DECREF((PyObject *)self);
        return NULL;
    }
    self->cd_last_empty = now;
    return (PyObject *)self;
}

static PyMethodDef CoDel_methods[] = {
    {"overloaded", CoDel_overloaded, METH_O, NULL},
    {"can_drop", (PyCFunction)(void (*)(void))CoDel_can_drop,
     METH_FASTCALL, NULL},
    {"get_drop_next", CoDel_get_drop_next, METH_O, NULL},
    {"empty", CoDel_empty, METH_NOARGS, NULL},
    {"get_max_idle", CoDel_get_max_idle, METH_NOARGS, NULL},
    {NULL, NULL, 0, NULL},
};

static PyMemberDef CoDel_members[] = {
    {(char *)"cd_targdelay", T_DOUBLE, offsetof(CoDelOb, cd_targdelay),
     READONLY, NULL},
    {(char *)"cd_first_above_time", T_DOUBLE,
     offsetof(CoDelOb, cd_first_above_time), 0, NULL},
    {(char *)"cd_drop_next", T_DOUBLE, offsetof(CoDelOb, cd_drop_next),
     0, NULL},
    {(char *)"cd_count", T_LONG, offsetof(CoDelOb, cd_count), 0, NULL},
    {(char *)"cd_last_empty", T_DOUBLE,
     offsetof(CoDelOb, cd_last_empty), 0, NULL},
    {NULL, 0, 0, 0, NULL},
};

static PyGetSetDef CoDel_getset[] = {
    {(char *)"cd_dropping", CoDel_get_dropping, NULL, NULL, NULL},
    {NULL, NULL, NULL, NULL, NULL},
};

PyTypeObject CoDelType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.ControlledDelay",
    sizeof(CoDelOb),
};

static void
codel_type_init(void)
{
    CoDelType.tp_dealloc = CoDel_dealloc;
    CoDelType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    CoDelType.tp_traverse = CoDel_traverse;
    CoDelType.tp_clear = CoDel_clear_;
    CoDelType.tp_methods = CoDel_methods;
    CoDelType.tp_members = CoDel_members;
    CoDelType.tp_getset = CoDel_getset;
    CoDelType.tp_new = CoDel_new;
}

/* ------------------------------------------------------------------ */
/* SlotDispatch: the pool's per-slot stateChanged listener             */
/* (pool._slot_state_changed) with the idle-feed and busy no-op fast   */
/* paths in C; every other state (and every unusual condition)         */
/* delegates to the Python method, which remains the full logic.       */

PyObject *s_p_backends;          /* "p_backends" */
PyObject *s_p_dead;              /* "p_dead" */
PyObject *s_p_initq_node;        /* "p_initq_node" */
PyObject *s_connected_to_backend;/* "connectedToBackend" */
PyObject *s_slot_state_changed;  /* "_slot_state_changed" */
PyObject *s_remove_node;         /* "remove" */
PyObject *s_rebalance;           /* "rebalance" */
PyObject *s_set_unwanted;        /* "set_unwanted" */

typedef struct {
    PyObject_HEAD
    PyObject *sd_pool;      /* owned */
    PyObject *sd_fsm;       /* the slot (FSMOb), owned */
    PyObject *sd_key;       /* owned */
    NQueue *sd_idleq;       /* owned (never replaced by the pool) */
    NQueue *sd_waiters;     /* owned */
    PyObject *sd_codel;     /* owned native ControlledDelay or NULL */
    int sd_has_codel;       /* codel configured but not native =>
                             * python path */
} SlotDispatch;

extern PyTypeObject SlotDispatchType;

static int
sd_fallback(SlotDispatch *d, PyObject *st)
{
    PyObject *r = PyObject_CallMethodObjArgs(
        d->sd_pool, s_slot_state_changed, d->sd_fsm, d->sd_key, st,
        NULL);
    if (r == NULL)
        return -1;
    Py_DECREF(r);
    return 0;
}

static PyObject *
SlotDispatch_call(PyObject *self_, PyObject *args, PyObject *kwds)
{
    SlotDispatch *d = (SlotDispatch *)self_;
    (void)kwds;
    if (PyTuple_GET_SIZE(args) != 1) {
        PyErr_SetString(PyExc_TypeError, "SlotDispatch(state)");
        return NULL;
    }
    PyObject *st = PyTuple_GET_ITEM(args, 0);
    FSMOb *fsm = (FSMOb *)d->sd_fsm;

    /* common preconditions for both fast paths */
    PyObject *initq_node = PyObject_GetAttr(d->sd_fsm, s_p_initq_node);
    if (initq_node == NULL)
        return NULL;
    int in_initq = (initq_node != Py_None);
    Py_DECREF(initq_node);

    if (!in_initq &&
        (st == s_busy_st ||
         (PyUnicode_Check(st) &&
          PyUnicode_Compare(st, s_busy_st) == 0))) {
        /* busy: with no pinger handle and no stale idleq node this
         * dispatch is a no-op */
        PyObject *idleq_node = PyObject_GetAttr(d->sd_fsm,
                                                s_p_idleq_node);
        if (idleq_node == NULL)
            return NULL;
        int linked = (idleq_node != Py_None);
        Py_DECREF(idleq_node);
        if (!linked) {
            PyObject *hdl = PyObject_GetAttr(d->sd_fsm, s_csf_handle);
            if (hdl == NULL)
                return NULL;
            int pinger = 0;
            if (hdl != Py_None && PyObject_TypeCheck(hdl, &CHType))
                pinger = ((CHOb *)hdl)->chb_pinger;
            Py_DECREF(hdl);
            if (!pinger)
                Py_RETURN_NONE;
        }
        if (sd_fallback(d, st) < 0)
            return NULL;
        Py_RETURN_NONE;
    }

    if (!in_initq && (!d->sd_has_codel || d->sd_codel != NULL) &&
        (st == s_idle ||
         (PyUnicode_Check(st) &&
          PyUnicode_Compare(st, s_idle) == 0))) {
        PyObject *dead = PyObject_GetAttr(d->sd_pool, s_p_dead);
        if (dead == NULL)
            return NULL;
        int isdead = PyDict_Check(dead) ?
            PyDict_Contains(dead, d->sd_key) : 1;
        Py_DECREF(dead);
        if (isdead != 0) {
            /* dead-backend recovery (or error): python path */
            if (isdead < 0)
                return NULL;
            if (sd_fallback(d, st) < 0)
                return NULL;
            Py_RETURN_NONE;
        }
        /* emit connectedToBackend (cheap when nobody listens) */
        PyObject *eargs[3] = {s_connected_to_backend, d->sd_key,
                              d->sd_fsm};
        if (emitter_emit_core((Emitter *)d->sd_pool, eargs[0],
                              eargs + 1, 2) < 0)
            return NULL;
        /* still idle? (stale events tolerated) */
        PyObject *fs = fsm->f_state;
        if (!(fs == s_idle ||
              (fs != NULL && PyUnicode_Compare(fs, s_idle) == 0))) {
            /* the event already emitted above; only the trailing
             * was-idle-now-isn't unlink branch can still apply */
            PyObject *idn = PyObject_GetAttr(d->sd_fsm,
                                             s_p_idleq_node);
            if (idn == NULL)
                return NULL;
            if (idn != Py_None) {
                PyObject *rr = PyObject_CallMethodObjArgs(
                    idn, s_remove_node, NULL);
                if (rr == NULL) {
                    Py_DECREF(idn);
                    return NULL;
                }
                Py_DECREF(rr);
                Py_DECREF(idn);
                if (PyObject_SetAttr(d->sd_fsm, s_p_idleq_node,
                                     Py_None) < 0)
                    return NULL;
                PyObject *rb = PyObject_CallMethodObjArgs(
                    d->sd_pool, s_rebalance, NULL);
                if (rb == NULL)
                    return NULL;
                Py_DECREF(rb);
            } else {
                Py_DECREF(idn);
            }
            Py_RETURN_NONE;
        }
        {
            PyObject *backends = PyObject_GetAttr(d->sd_pool,
                                                  s_p_backends);
            if (backends == NULL)
                return NULL;
            int wanted = PyDict_Check(backends) ?
                PyDict_Contains(backends, d->sd_key) : 0;
            Py_DECREF(backends);
            if (wanted < 0)
                return NULL;
            if (!wanted) {
                /* no longer a backend: set_unwanted, like the python
                 * dispatcher (it has not emitted again: call the slot
                 * method directly) */
                PyObject *su = PyObject_CallMethodObjArgs(
                    d->sd_fsm, s_set_unwanted, NULL);
                if (su == NULL)
                    return NULL;
                Py_DECREF(su);
                Py_RETURN_NONE;
            }
        }
        /* feed waiters (with the CoDel drop check on each; like the
         * python dispatcher, overloaded() is fed BEFORE the staleness
         * check so the controller sees every dequeue) */
        while (d->sd_waiters->q_len > 0) {
            PyObject *hdl = nqueue_shift_value(d->sd_waiters);
            if (hdl == NULL)
                return NULL;
            if (!PyObject_TypeCheck(hdl, &CHType)) {
                Py_DECREF(hdl);
                continue;
            }
            int drop = 0;
            if (d->sd_codel != NULL) {
                drop = codel_overloaded_c((CoDelOb *)d->sd_codel,
                                          ((CHOb *)hdl)->chb_started);
                if (drop < 0) {
                    Py_DECREF(hdl);
                    return NULL;
                }
            }
            if (!ch_state_is((CHOb *)hdl, s_waiting)) {
                Py_DECREF(hdl);
                continue;
            }
            if (drop) {
                PyObject *r = CH_timeout(hdl, NULL);
                Py_DECREF(hdl);
                if (r == NULL)
                    return NULL;
                Py_DECREF(r);
                continue;
            }
            PyObject *r = CH_try_(hdl, d->sd_fsm);
            Py_DECREF(hdl);
            if (r == NULL)
                return NULL;
            Py_DECREF(r);
            Py_RETURN_NONE;
        }
        if (d->sd_codel != NULL &&
            codel_empty_c((CoDelOb *)d->sd_codel) < 0)
            return NULL;
        /* no waiter: park on the idle queue */
        QNode *n = nqueue_push(d->sd_idleq, d->sd_fsm);
        if (n == NULL)
            return NULL;
        int rc = PyObject_SetAttr(d->sd_fsm, s_p_idleq_node,
                                  (PyObject *)n);
        Py_DECREF((PyObject *)n);
        if (rc < 0)
            return NULL;
        Py_RETURN_NONE;
    }

    if (sd_fallback(d, st) < 0)
        return NULL;
    Py_RETURN_NONE;
}

static int
SlotDispatch_traverse(PyObject *self_, visitproc visit, void *arg)
{
    SlotDispatch *d = (SlotDispatch *)self_;
    Py_VISIT(d->sd_pool);
    Py_VISIT(d->sd_fsm);
    Py_VISIT(d->sd_key);
    Py_VISIT((PyObject *)d->sd_idleq);
    Py_VISIT((PyObject *)d->sd_waiters);
    Py_VISIT(d->sd_codel);
    return 0;
}

static int
SlotDispatch_clear_(PyObject *self_)
{
    SlotDispatch *d = (SlotDispatch *)self_;
    Py_CLEAR(d->sd_pool);
    Py_CLEAR(d->sd_fsm);
    Py_CLEAR(d->sd_key);
    Py_CLEAR(d->sd_idleq);
    Py_CLEAR(d->sd_waiters);
    Py_CLEAR(d->sd_codel);
    return 0;
}

static void
SlotDispatch_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    SlotDispatch_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyObject *
SlotDispatch_new(PyTypeObject *type, PyObject *args, PyObject *kwds)
{
    PyObject *pool, *fsm, *key, *has_codel;
    (void)kwds;
    if (!PyArg_ParseTuple(args, "OOOO", &pool, &fsm, &key, &has_codel))
        return NULL;
    if (!PyObject_TypeCheck(pool, &EmitterType) ||
        !PyObject_TypeCheck(fsm, &FSMType)) {
        PyErr_SetString(PyExc_TypeError,
                        "SlotDispatch(pool, slot, key, has_codel)");
        return NULL;
    }
    PyObject *iq = PyObject_GetAttrString(pool, "p_idleq");
    PyObject *wq = iq ? PyObject_GetAttrString(pool, "p_waiters") : NULL;
    if (wq == NULL || !PyObject_TypeCheck(iq, &NQueueType) ||
        !PyObject_TypeCheck(wq, &NQueueType)) {
        Py_XDECREF(iq);
        Py_XDECREF(wq);
        if (!PyErr_Occurred())
            PyErr_SetString(PyExc_TypeError,
                            "pool queues must be native");
        return NULL;
    }
    SlotDispatch *d = PyObject_GC_New(SlotDispatch, type);
    if (d == NULL) {
        Py_DECREF(iq);
        Py_DECREF(wq);
        return NULL;
    }
    Py_INCREF(pool);
    d->sd_pool = pool;
    Py_INCREF(fsm);
    d->sd_fsm = fsm;
    Py_INCREF(key);
    d->sd_key = key;
    d->sd_idleq = (NQueue *)iq;
    d->sd_waiters = (NQueue *)wq;
    d->sd_has_codel = PyObject_IsTrue(has_codel);
    d->sd_codel = NULL;
    PyObject_GC_Track((PyObject *)d);
    if (d->sd_has_codel) {
        PyObject *cd = PyObject_GetAttrString(pool, "p_codel");
        if (cd == NULL) {
            PyErr_Clear();
        } else if (PyObject_TypeCheck(cd, &CoDelType)) {
            d->sd_codel = cd;     /* native: feed path stays in C */
        } else {
            Py_DECREF(cd);        /* pure codel: python feed path */
        }
    }
    return (PyObject *)d;
}

PyTypeObject SlotDispatchType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.SlotDispatch",
    sizeof(SlotDispatch),
};

/* -- SlotKit type ---------------------------------------------------- */

static int
SlotKit_traverse(PyObject *self_, visitproc visit, void *arg)
{
    SlotKit *k = (SlotKit *)self_;
    Py_VISIT((PyObject *)k->k_slot);
    Py_VISIT(k->k_smgr);
    Py_VISIT(k->k_observed);
    Py_VISIT((PyObject *)k->k_busy_scope);
    Py_VISIT((PyObject *)k->k_idle_scope);
    for (int i = 0; i < 4; i++)
        Py_VISIT(k->k_cbs[i]);
    return 0;
}

static int
SlotKit_clear_(PyObject *self_)
{
    SlotKit *k = (SlotKit *)self_;
    Py_CLEAR(k->k_slot);
    Py_CLEAR(k->k_smgr);
    Py_CLEAR(k->k_observed);
    Py_CLEAR(k->k_busy_scope);
    Py_CLEAR(k->k_idle_scope);
    for (int i = 0; i < 4; i++)
        Py_CLEAR(k->k_cbs[i]);
    return 0;
}

static void
SlotKit_dealloc(PyObject *self_)
{
    PyObject_GC_UnTrack(self_);
    SlotKit_clear_(self_);
    PyObject_GC_Del(self_);
}

static PyObject *
SlotKit_new(PyTypeObject *type, PyObject *args, PyObject *kwds)
{
    PyObject *slot, *smgr;
    (void)kwds;
    if (!PyArg_ParseTuple(args, "OO", &slot, &smgr))
        return NULL;
    if (!PyObject_TypeCheck(slot, &FSMType) ||
        !PyObject_TypeCheck(smgr, &FSMType)) {
        PyErr_SetString(PyExc_TypeError,
                        "SlotKit(slot, smgr): native FSMs required");
        return NULL;
    }
    SlotKit *k = PyObject_GC_New(SlotKit, type);
    if (k == NULL)
        return NULL;
    Py_INCREF(slot);
    k->k_slot = (FSMOb *)slot;
    Py_INCREF(smgr);
    k->k_smgr = smgr;
    k->k_observed = NULL;
    k->k_busy_scope = NULL;
    k->k_idle_scope = NULL;
    for (int i = 0; i < 4; i++)
        k->k_cbs[i] = NULL;
    PyObject_GC_Track((PyObject *)k);
    for (int i = 0; i < 4; i++) {
        KitCb *cb = PyObject_GC_New(KitCb, &KitCbType);
        if (cb == NULL) {
            Py_DECREF((PyObject *)k);
            return NULL;
        }
        Py_INCREF((PyObject *)k);
        cb->kc_kit = k;
        cb->kc_which = i;
        PyObject_GC_Track((PyObject *)cb);
        k->k_cbs[i] = (PyObject *)cb;
    }
    return (PyObject *)k;
}

PyTypeObject SlotKitType = {
    PyVarObject_HEAD_INIT(NULL, 0)
    "cueball_amd._speed.SlotKit",
    sizeof(SlotKit),
};

static void
slotkit_types_init(void)
{
    KitCbType.tp_dealloc = KitCb_dealloc;
    KitCbType.tp_call = KitCb_call;
    KitCbType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    KitCbType.tp_traverse = KitCb_traverse;
    KitCbType.tp_clear = KitCb_clear_;

    SlotKitType.tp_dealloc = SlotKit_dealloc;
    SlotKitType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    SlotKitType.tp_traverse = SlotKit_traverse;
    SlotKitType.tp_clear = SlotKit_clear_;
    SlotKitType.tp_new = SlotKit_new;

    SlotDispatchType.tp_dealloc = SlotDispatch_dealloc;
    SlotDispatchType.tp_call = SlotDispatch_call;
    SlotDispatchType.tp_flags = Py_TPFLAGS_DEFAULT | Py_TPFLAGS_HAVE_GC;
    SlotDispatchType.tp_traverse = SlotDispatch_traverse;
    SlotDispatchType.tp_clear = SlotDispatch_clear_;
    SlotDispatchType.tp_new = SlotDispatch_new;
}

PyObject *
speed_set_tracer(PyObject *mod, PyObject *fn)
{
    (void)mod;
    if (fn == Py_None) {
        Py_CLEAR(g_tracer);
    } else {
        Py_INCREF(fn);
        Py_XSETREF(g_tracer, fn);
    }
    Py_RETURN_NONE;
}

PyMethodDef speed_methods[] = {
    {"_set_claim_helpers",
     (PyCFunction)(void (*)(void))speed_set_claim_helpers, METH_FASTCALL,
     NULL},
    {"_set_tracer", speed_set_tracer, METH_O, NULL},
    {"count_listeners",
     (PyCFunction)(void (*)(void))speed_count_listeners, METH_FASTCALL,
     NULL},
    {"_set_helpers", (PyCFunction)(void (*)(void))speed_set_helpers,
     METH_FASTCALL, NULL},
    {"claim_fast", (PyCFunction)(void (*)(void))speed_claim_fast,
     METH_FASTCALL, NULL},
    {"pool_claim", (PyCFunction)(void (*)(void))speed_pool_claim,
     METH_FASTCALL, NULL},
    {NULL, NULL, 0, NULL},
};

struct PyModuleDef speedmodule = {
    PyModuleDef_HEAD_INIT,
    "cueball_amd._speed",
    "Native event/FSM runtime core",
    -1,
    speed_methods,
};

}  /* namespace */

PyMODINIT_FUNC
PyInit__speed(void)
{
    s_stateChanged = PyUnicode_InternFromString("stateChanged");
    s_listener = PyUnicode_InternFromString("listener");
    s_on = PyUnicode_InternFromString("on");
    s_remove_listener = PyUnicode_InternFromString("remove_listener");
    s_call_soon = PyUnicode_InternFromString("call_soon");
    s_call_later = PyUnicode_InternFromString("call_later");
    s_cancel = PyUnicode_InternFromString("cancel");
    s_state_prefix = PyUnicode_InternFromString("state_");
    s_dot = PyUnicode_InternFromString(".");
    s_underscore = PyUnicode_InternFromString("_");
    s_flush_name = PyUnicode_InternFromString("_flush_state_changed");
    s_internal = PyUnicode_InternFromString("_cueball_internal");
    s_is_closed = PyUnicode_InternFromString("is_closed");
    s_claim = PyUnicode_InternFromString("claim");
    s_warn = PyUnicode_InternFromString("warn");
    s_incr = PyUnicode_InternFromString("_incr_counter");
    s_claim_timeout_evt = PyUnicode_InternFromString("claim-timeout");
    s_is_in_state = PyUnicode_InternFromString("is_in_state");
    s_time = PyUnicode_InternFromString("time");
    s_waiting = PyUnicode_InternFromString("waiting");
    s_claiming = PyUnicode_InternFromString("claiming");
    s_claimed = PyUnicode_InternFromString("claimed");
    s_released = PyUnicode_InternFromString("released");
    s_closed = PyUnicode_InternFromString("closed");
    s_cancelled = PyUnicode_InternFromString("cancelled");
    s_failed = PyUnicode_InternFromString("failed");
    s_idle = PyUnicode_InternFromString("idle");
    s_error_evt = PyUnicode_InternFromString("error");
    t_waiting_valid = PyTuple_Pack(3, s_claiming, s_cancelled, s_failed);
    t_claiming_valid = PyTuple_Pack(3, s_claimed, s_waiting, s_cancelled);
    t_claimed_valid = PyTuple_Pack(2, s_released, s_closed);
    t_empty = PyTuple_New(0);
    s_leak_events[0] = PyUnicode_InternFromString("close");
    s_leak_events[1] = s_error_evt;
    s_leak_events[2] = PyUnicode_InternFromString("readable");
    s_leak_events[3] = PyUnicode_InternFromString("data");
    g_entry_name_cache = PyDict_New();
    if (g_entry_name_cache == NULL)
        return NULL;
    g_flush_batches = PyDict_New();
    if (g_flush_batches == NULL)
        return NULL;

    s_running_st = PyUnicode_InternFromString("running");
    s_p_idleq_node = PyUnicode_InternFromString("p_idleq_node");
    s_p_total_conns = PyUnicode_InternFromString("p_total_conns");
    s_p_busy_hwm = PyUnicode_InternFromString("p_busy_hwm");
    s_p_demand_hwm = PyUnicode_InternFromString("p_demand_hwm");
    s_ticket_slow = PyUnicode_InternFromString("_ticket_slow");
    s_ch_waiter_node = PyUnicode_InternFromString("ch_waiter_node");
    s_p_counters = PyUnicode_InternFromString("p_counters");
    s_p_rebal_scheduled = PyUnicode_InternFromString("p_rebal_scheduled");
    s_max_claim_queue = PyUnicode_InternFromString("max-claim-queue");
    s_queued_claim = PyUnicode_InternFromString("queued-claim");
    s_connected_st = PyUnicode_InternFromString("connected");
    s_busy_st = PyUnicode_InternFromString("busy");
    s_stopping_st = PyUnicode_InternFromString("stopping");
    s_stopped_st = PyUnicode_InternFromString("stopped");
    s_retrying_st = PyUnicode_InternFromString("retrying");
    s_connecting_st = PyUnicode_InternFromString("connecting");
    s_killing_st = PyUnicode_InternFromString("killing");
    s_unwanted_evt = PyUnicode_InternFromString("unwanted");
    s_csf_handle = PyUnicode_InternFromString("csf_handle");
    s_csf_prev_handle = PyUnicode_InternFromString("csf_prev_handle");
    s_csf_wanted = PyUnicode_InternFromString("csf_wanted");
    s_csf_monitor = PyUnicode_InternFromString("csf_monitor");
    s_sm_socket = PyUnicode_InternFromString("sm_socket");
    t_busy_valid = PyTuple_Pack(6, s_idle, s_stopping_st, s_stopped_st,
                                s_retrying_st, s_killing_st,
                                s_connecting_st);
    t_idle_valid = PyTuple_Pack(5, s_retrying_st, s_connecting_st,
                                s_stopping_st, s_stopped_st, s_busy_st);
    s_p_backends = PyUnicode_InternFromString("p_backends");
    s_p_dead = PyUnicode_InternFromString("p_dead");
    s_p_initq_node = PyUnicode_InternFromString("p_initq_node");
    s_connected_to_backend =
        PyUnicode_InternFromString("connectedToBackend");
    s_slot_state_changed =
        PyUnicode_InternFromString("_slot_state_changed");
    s_remove_node = PyUnicode_InternFromString("remove");
    s_rebalance = PyUnicode_InternFromString("rebalance");
    s_set_unwanted = PyUnicode_InternFromString("set_unwanted");
    queue_types_init();
    ct_type_init();
    slotkit_types_init();
    codel_type_init();
    if (PyType_Ready(&EmitterType) < 0 ||
        PyType_Ready(&OnceWrapperType) < 0 ||
        PyType_Ready(&GuardedCbType) < 0 ||
        PyType_Ready(&ScopeType) < 0 ||
        PyType_Ready(&IntervalType) < 0 ||
        PyType_Ready(&FlushBatchType) < 0 ||
        PyType_Ready(&ConnErrCbType) < 0 ||
        PyType_Ready(&FailCbType) < 0 ||
        PyType_Ready(&CHType) < 0 ||
        PyType_Ready(&FSMType) < 0 ||
        PyType_Ready(&QNodeType) < 0 ||
        PyType_Ready(&NQueueType) < 0 ||
        PyType_Ready(&CTType) < 0 ||
        PyType_Ready(&KitCbType) < 0 ||
        PyType_Ready(&SlotKitType) < 0 ||
        PyType_Ready(&SlotDispatchType) < 0 ||
        PyType_Ready(&CoDelType) < 0)
        return NULL;

    g_remove_desc = PyDict_GetItemString(EmitterType.tp_dict,
                                         "remove_listener");
    if (g_remove_desc == NULL)
        return NULL;
    Py_INCREF(g_remove_desc);

    PyObject *m = PyModule_Create(&speedmodule);
    if (m == NULL)
        return NULL;
    Py_INCREF(&EmitterType);
    PyModule_AddObject(m, "EventEmitter", (PyObject *)&EmitterType);
    Py_INCREF(&ScopeType);
    PyModule_AddObject(m, "StateScope", (PyObject *)&ScopeType);
    Py_INCREF(&FSMType);
    PyModule_AddObject(m, "FSM", (PyObject *)&FSMType);
    Py_INCREF(&CHType);
    PyModule_AddObject(m, "ClaimHandleBase", (PyObject *)&CHType);
    Py_INCREF(&NQueueType);
    PyModule_AddObject(m, "Queue", (PyObject *)&NQueueType);
    Py_INCREF(&QNodeType);
    PyModule_AddObject(m, "QueueNode", (PyObject *)&QNodeType);
    Py_INCREF(&CTType);
    PyModule_AddObject(m, "ClaimTicket", (PyObject *)&CTType);
    Py_INCREF(&SlotKitType);
    PyModule_AddObject(m, "SlotKit", (PyObject *)&SlotKitType);
    Py_INCREF(&SlotDispatchType);
    PyModule_AddObject(m, "SlotDispatch", (PyObject *)&SlotDispatchType);
    Py_INCREF(&CoDelType);
    PyModule_AddObject(m, "ControlledDelay", (PyObject *)&CoDelType);
    return m;
}
