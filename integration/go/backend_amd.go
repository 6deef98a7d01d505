// backend_amd.go — complete cgo Backend over the MI355X slab C-ABI.
//
// Drop this file (plus include/kb_slab.h and the built libkbslab.so) into
// pkg/backend of kubewharf/kubebrain and construct with NewAmdBackend in
// place of backend.NewBackend (cmd wiring: option.Run). Every method of the
// Backend interface (pkg/backend/backend.go:44-84) maps 1:1 onto a kb_*
// entry point with the same argument meaning, revision arithmetic and error
// taxonomy (include/kb_slab.h cites the replaced reference code per symbol).
//
// NOTE: the build image this repo ships from has no Go toolchain
// (SURVEY.md §8c), so this file is delivered unreviewed-by-the-compiler;
// the same call sequences are exercised end-to-end by the repo's ctypes
// harness (kubebrain_amd/client.py + tests/test_gpu_parity.py), which is
// the executable specification of the wire formats decoded below.
package backend

/*
#cgo CFLAGS: -I${SRCDIR}/../../include
#cgo LDFLAGS: -L${SRCDIR}/../../lib -lkbslab
#include <stdlib.h>
#include "kb_slab.h"
*/
import "C"

import (
	"context"
	"encoding/binary"
	"fmt"
	"time"
	"unsafe"

	proto "github.com/kubewharf/kubebrain-client/api/v2rpc"
	"k8s.io/client-go/tools/leaderelection/resourcelock"

	"github.com/kubewharf/kubebrain/pkg/storage"
)

type amdBackend struct {
	h    *C.kb_store
	lock resourcelock.Interface // same resource lock the stock backend wires
}

// NewAmdBackend opens the GPU slab store. prefix/watchCacheSize/eventsTTL
// carry backend.Config's meaning (backend.go:86-120).
func NewAmdBackend(prefix string, watchCacheSize int, eventsTTL time.Duration,
	etcdCompat bool, lock resourcelock.Interface) (Backend, error) {
	cp := C.CString(prefix)
	defer C.free(unsafe.Pointer(cp))
	compat := C.int(0)
	if etcdCompat {
		compat = 1
	}
	h := C.kb_new(cp, C.int(watchCacheSize),
		C.longlong(int64(eventsTTL/time.Second)), compat)
	if h == nil {
		return nil, lastErr("kb_new")
	}
	return &amdBackend{h: h, lock: lock}, nil
}

func (b *amdBackend) Close() { C.kb_free(b.h) }

// ---- error taxonomy (storage/interface.go:140-147 <-> kb_status) ----

func lastErr(op string) error {
	var buf [512]C.char
	code := C.kb_last_error(&buf[0], 512)
	return fmt.Errorf("%s: kb_status %d: %s", op, int(code),
		C.GoString(&buf[0]))
}

func statusToErr(rc C.int) error {
	switch rc {
	case C.KB_OK:
		return nil
	case C.KB_ENOTFOUND:
		return storage.ErrKeyNotFound
	case C.KB_ECAS:
		return storage.ErrCASFailed
	case C.KB_ECOMPACTED:
		return fmt.Errorf("required revision has been compacted")
	case C.KB_EREVDRIFT:
		return storage.ErrRevisionDriftBack
	case C.KB_EKEYTOOLONG, C.KB_EBADKEY, C.KB_EINVALID:
		return fmt.Errorf("invalid argument (kb_status %d)", int(rc))
	default:
		return fmt.Errorf("kb_status %d", int(rc))
	}
}

func bptr(b []byte) *C.uint8_t {
	if len(b) == 0 {
		return nil
	}
	return (*C.uint8_t)(unsafe.Pointer(&b[0]))
}

func header(rev uint64) *proto.ResponseHeader {
	return &proto.ResponseHeader{Revision: rev}
}

// ---- writes (txn.go:33-265 protocol, executed inside libkbslab) ----

func (b *amdBackend) Create(ctx context.Context, r *proto.CreateRequest) (*proto.CreateResponse, error) {
	var rev C.uint64_t
	var ok C.int
	rc := C.kb_create(b.h, bptr(r.Key), C.size_t(len(r.Key)),
		bptr(r.Value), C.size_t(len(r.Value)), &rev, &ok)
	if rc != C.KB_OK && rc != C.KB_ECAS {
		return nil, statusToErr(rc)
	}
	return &proto.CreateResponse{Header: header(uint64(rev)),
		Succeeded: ok != 0}, nil
}

func (b *amdBackend) Update(ctx context.Context, r *proto.UpdateRequest) (*proto.UpdateResponse, error) {
	kv := r.GetKv()
	val := make([]byte, 1<<20)
	var rev, kvRev C.uint64_t
	var ok, hasKv C.int
	var kvLen C.size_t
	rc := C.kb_update(b.h, bptr(kv.Key), C.size_t(len(kv.Key)),
		bptr(kv.Value), C.size_t(len(kv.Value)), C.uint64_t(kv.Revision),
		&rev, &ok, &hasKv, bptr(val), C.size_t(len(val)), &kvLen, &kvRev)
	if rc != C.KB_OK && rc != C.KB_ECAS {
		return nil, statusToErr(rc)
	}
	resp := &proto.UpdateResponse{Header: header(uint64(rev)),
		Succeeded: ok != 0}
	if hasKv != 0 { // CAS failure returns the latest kv (txn.go:221-239)
		resp.Kv = &proto.KeyValue{Key: kv.Key,
			Value: append([]byte(nil), val[:kvLen]...),
			Revision: uint64(kvRev)}
	}
	return resp, nil
}

func (b *amdBackend) Delete(ctx context.Context, r *proto.DeleteRequest) (*proto.DeleteResponse, error) {
	val := make([]byte, 1<<20)
	var rev, kvRev C.uint64_t
	var ok, hasKv C.int
	var kvLen C.size_t
	rc := C.kb_delete(b.h, bptr(r.Key), C.size_t(len(r.Key)),
		C.uint64_t(r.Revision), &rev, &ok, &hasKv, bptr(val),
		C.size_t(len(val)), &kvLen, &kvRev)
	if rc != C.KB_OK && rc != C.KB_ECAS {
		return nil, statusToErr(rc)
	}
	resp := &proto.DeleteResponse{Header: header(uint64(rev)),
		Succeeded: ok != 0}
	if hasKv != 0 {
		resp.Kv = &proto.KeyValue{Key: r.Key,
			Value: append([]byte(nil), val[:kvLen]...),
			Revision: uint64(kvRev)}
	}
	return resp, nil
}

// ---- reads (the GPU hot path: range.go:34-205 semantics) ----

func (b *amdBackend) Get(ctx context.Context, r *proto.GetRequest) (*proto.GetResponse, error) {
	val := make([]byte, 4<<20)
	var hdr, modRev C.uint64_t
	var hasKv C.int
	var vlen C.size_t
	rc := C.kb_get(b.h, bptr(r.Key), C.size_t(len(r.Key)),
		C.uint64_t(r.Revision), &hdr, &hasKv, bptr(val), C.size_t(len(val)),
		&vlen, &modRev)
	if rc == C.KB_ENOTFOUND {
		return &proto.GetResponse{Header: header(uint64(hdr))}, nil
	}
	if rc != C.KB_OK {
		return nil, statusToErr(rc)
	}
	resp := &proto.GetResponse{Header: header(uint64(hdr))}
	if hasKv != 0 {
		resp.Kv = &proto.KeyValue{Key: r.Key,
			Value: append([]byte(nil), val[:vlen]...),
			Revision: uint64(modRev)}
	}
	return resp, nil
}

// decodeKvs parses kb_list/kb_stream_next's wire format
// {u32 n; n x {u64 rev; u32 klen; key; u32 vlen; val}} (little-endian;
// executable spec: tests/kbclient.py _parse_kvs).
func decodeKvs(buf []byte) []*proto.KeyValue {
	if len(buf) < 4 {
		return nil
	}
	n := binary.LittleEndian.Uint32(buf)
	off := 4
	out := make([]*proto.KeyValue, 0, n)
	for i := uint32(0); i < n; i++ {
		rev := binary.LittleEndian.Uint64(buf[off:])
		klen := binary.LittleEndian.Uint32(buf[off+8:])
		off += 12
		key := append([]byte(nil), buf[off:off+int(klen)]...)
		off += int(klen)
		vlen := binary.LittleEndian.Uint32(buf[off:])
		off += 4
		val := append([]byte(nil), buf[off:off+int(vlen)]...)
		off += int(vlen)
		out = append(out, &proto.KeyValue{Key: key, Value: val, Revision: rev})
	}
	return out
}

func (b *amdBackend) List(ctx context.Context, r *proto.RangeRequest) (*proto.RangeResponse, error) {
	out := make([]byte, 16<<20)
	var outLen C.size_t
	var hdr C.uint64_t
	var more C.int
	for {
		rc := C.kb_list(b.h, bptr(r.Key), C.size_t(len(r.Key)),
			bptr(r.End), C.size_t(len(r.End)), C.uint64_t(r.Revision),
			C.int64_t(r.Limit), bptr(out), C.size_t(len(out)), &outLen,
			&hdr, &more)
		if rc == C.KB_ENOBUF { // grow and retry; out_len carries need
			out = make([]byte, int(outLen))
			continue
		}
		if rc != C.KB_OK {
			return nil, statusToErr(rc)
		}
		break
	}
	return &proto.RangeResponse{Header: header(uint64(hdr)),
		Kvs: decodeKvs(out[:outLen]), More: more != 0}, nil
}

func (b *amdBackend) Count(ctx context.Context, r *proto.CountRequest) (*proto.CountResponse, error) {
	var hdr, cnt C.uint64_t
	rc := C.kb_count(b.h, bptr(r.Key), C.size_t(len(r.Key)),
		bptr(r.End), C.size_t(len(r.End)), &hdr, &cnt)
	if rc != C.KB_OK {
		return nil, statusToErr(rc)
	}
	return &proto.CountResponse{Header: header(uint64(hdr)),
		Count: uint64(cnt)}, nil
}

func (b *amdBackend) Compact(ctx context.Context, revision uint64) (*proto.CompactResponse, error) {
	var outRev C.uint64_t
	rc := C.kb_compact(b.h, C.uint64_t(revision), &outRev)
	if rc != C.KB_OK {
		return nil, statusToErr(rc)
	}
	return &proto.CompactResponse{Header: header(uint64(outRev))}, nil
}

// ---- partitions + streaming (range.go:208-256, batches of 300) ----

func (b *amdBackend) GetPartitions(ctx context.Context, r *proto.ListPartitionRequest) (*proto.ListPartitionResponse, error) {
	out := make([]byte, 64<<10)
	var outLen C.size_t
	var hdr C.uint64_t
	rc := C.kb_partitions(b.h, bptr(r.Key), C.size_t(len(r.Key)),
		bptr(r.End), C.size_t(len(r.End)), bptr(out), C.size_t(len(out)),
		&outLen, &hdr)
	if rc != C.KB_OK {
		return nil, statusToErr(rc)
	}
	// wire: u32 n; n x {u32 len; bytes} partition border keys
	n := binary.LittleEndian.Uint32(out)
	off := 4
	borders := make([][]byte, 0, n)
	for i := uint32(0); i < n; i++ {
		l := binary.LittleEndian.Uint32(out[off:])
		off += 4
		borders = append(borders, append([]byte(nil), out[off:off+int(l)]...))
		off += int(l)
	}
	return &proto.ListPartitionResponse{Header: header(uint64(hdr)),
		PartitionNum: int64(len(borders) - 1), PartitionKeys: borders}, nil
}

func (b *amdBackend) ListByStream(ctx context.Context, startKey, endKey []byte, revision uint64) (<-chan *proto.StreamRangeResponse, error) {
	var readRev C.uint64_t
	var st C.int
	sid := C.kb_stream_open(b.h, bptr(startKey), C.size_t(len(startKey)),
		bptr(endKey), C.size_t(len(endKey)), C.uint64_t(revision), &readRev,
		&st)
	if st != C.KB_OK {
		return nil, statusToErr(st)
	}
	ch := make(chan *proto.StreamRangeResponse, 4)
	go func() {
		defer close(ch)
		defer C.kb_stream_close(b.h, sid)
		buf := make([]byte, 16<<20)
		for {
			var outLen C.size_t
			rc := C.kb_stream_next(b.h, sid, bptr(buf), C.size_t(len(buf)),
				&outLen)
			if rc == C.KB_ENOBUF {
				buf = make([]byte, int(outLen))
				continue
			}
			if rc != C.KB_OK {
				ch <- &proto.StreamRangeResponse{Err: statusToErr(rc).Error()}
				return
			}
			kvs := decodeKvs(buf[:outLen])
			if len(kvs) == 0 { // empty batch = end marker
				return
			}
			select {
			case ch <- &proto.StreamRangeResponse{
				RangeResponse: &proto.RangeResponse{
					Header: header(uint64(readRev)), Kvs: kvs}}:
			case <-ctx.Done():
				return
			}
		}
	}()
	return ch, nil
}

// ---- watch (watch.go:37-159; poller goroutine over kb_watch_poll) ----

func (b *amdBackend) Watch(ctx context.Context, key string, revision uint64) (<-chan []*proto.Event, error) {
	var st C.int
	kb := []byte(key)
	wid := C.kb_watch(b.h, bptr(kb), C.size_t(len(kb)),
		C.uint64_t(revision), &st)
	if st != C.KB_OK {
		return nil, statusToErr(st)
	}
	ch := make(chan []*proto.Event, 16)
	go func() {
		defer close(ch)
		defer C.kb_watch_cancel(b.h, wid)
		buf := make([]byte, 4<<20)
		tick := time.NewTicker(time.Millisecond)
		defer tick.Stop()
		for {
			select {
			case <-ctx.Done():
				return
			case <-tick.C:
			}
			var outLen C.size_t
			rc := C.kb_watch_poll(b.h, wid, bptr(buf), C.size_t(len(buf)),
				&outLen)
			if rc == C.KB_ENOBUF { // queue intact; retry larger
				buf = make([]byte, int(outLen))
				continue
			}
			if rc != C.KB_OK { // incl. KB_EWATCH_DROPPED (slow consumer)
				return
			}
			if evs := decodeEvents(buf[:outLen]); len(evs) > 0 {
				select {
				case ch <- evs:
				case <-ctx.Done():
					return
				}
			}
		}
	}()
	return ch, nil
}

// decodeEvents parses kb_watch_poll's wire format {u32 n; n x {i32 type;
// u64 rev; u64 kv_rev; u32 klen; key; u32 vlen; val}}
// (executable spec: tests/kbclient.py _parse_events).
func decodeEvents(buf []byte) []*proto.Event {
	if len(buf) < 4 {
		return nil
	}
	n := binary.LittleEndian.Uint32(buf)
	off := 4
	out := make([]*proto.Event, 0, n)
	for i := uint32(0); i < n; i++ {
		typ := int32(binary.LittleEndian.Uint32(buf[off:]))
		rev := binary.LittleEndian.Uint64(buf[off+4:])
		kvRev := binary.LittleEndian.Uint64(buf[off+12:])
		klen := binary.LittleEndian.Uint32(buf[off+20:])
		off += 24
		key := append([]byte(nil), buf[off:off+int(klen)]...)
		off += int(klen)
		vlen := binary.LittleEndian.Uint32(buf[off:])
		off += 4
		val := append([]byte(nil), buf[off:off+int(vlen)]...)
		off += int(vlen)
		_ = rev
		out = append(out, &proto.Event{
			Type: proto.Event_EventType(typ),
			Kv:   &proto.KeyValue{Key: key, Value: val, Revision: kvRev},
		})
	}
	return out
}

// ---- TSO + election ----

func (b *amdBackend) GetResourceLock() resourcelock.Interface { return b.lock }

func (b *amdBackend) GetCurrentRevision() uint64 {
	return uint64(C.kb_current_rev(b.h))
}

func (b *amdBackend) SetCurrentRevision(rev uint64) {
	C.kb_set_current_rev(b.h, C.ulonglong(rev))
}
