/* jsprim shim: the helpers the reference tests use. */
'use strict';

module.exports = {
	forEachKey: function (obj, fn) {
		Object.keys(obj).forEach(function (k) {
			fn(k, obj[k]);
		});
	},
	deepCopy: function (v) {
		return (JSON.parse(JSON.stringify(v)));
	},
	isEmpty: function (obj) {
		return (Object.keys(obj).length === 0);
	}
};
