/*
 * sinon shim (1.x API surface the reference tests use):
 * sinon.sandbox.create() -> { stub(obj, name, replacement), restore() }.
 */

'use strict';

function createSandbox() {
	const saved = [];
	return ({
		stub: function (obj, name, replacement) {
			saved.push([obj, name, obj[name],
			    Object.prototype.hasOwnProperty.call(obj, name)]);
			if (replacement === undefined) {
				replacement = function () { };
			}
			obj[name] = replacement;
			return (replacement);
		},
		restore: function () {
			while (saved.length > 0) {
				const s = saved.pop();
				if (s[3])
					s[0][s[1]] = s[2];
				else
					delete s[0][s[1]];
			}
		}
	});
}

module.exports = {
	sandbox: { create: createSandbox },
	createSandbox: createSandbox
};
